from .logging import logger
from .timing import StepTimer, barrier_sync

__all__ = ["logger", "StepTimer", "barrier_sync"]
