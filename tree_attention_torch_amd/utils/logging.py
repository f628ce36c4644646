"""Rank-tagged logging with a loguru-compatible surface.

The reference (/root/reference/model.py:5,160) uses loguru with a rotating
file sink (``logger.add("tree_attention_log.log", rotation="10 MB")``).
loguru is not guaranteed to be installed in deployment images, so this module
provides the same surface (``logger.info/debug/warning/error/add/remove``)
backed by the stdlib, and transparently uses the real loguru when available.

Every line is tagged with the distributed rank (read lazily from the
``RANK`` env var or torch.distributed) so per-rank logs interleave readably.
"""

from __future__ import annotations

import logging
import logging.handlers
import os
import sys
from typing import Any

try:  # pragma: no cover - exercised only when loguru is installed
    from loguru import logger as _loguru_logger

    HAS_LOGURU = True
except ImportError:
    _loguru_logger = None
    HAS_LOGURU = False


def _current_rank() -> str:
    r = os.environ.get("RANK")
    if r is not None:
        return r
    try:
        import torch.distributed as dist

        if dist.is_available() and dist.is_initialized():
            return str(dist.get_rank())
    except Exception:
        pass
    return "-"


class _RankFilter(logging.Filter):
    def filter(self, record: logging.LogRecord) -> bool:
        record.rank = _current_rank()
        return True


_FMT = "%(asctime)s | %(levelname)-7s | rank=%(rank)s | %(message)s"


class _ShimLogger:
    """Stdlib-backed logger exposing the loguru calls the framework uses."""

    def __init__(self) -> None:
        self._log = logging.getLogger("tree_attention")
        self._log.setLevel(logging.DEBUG)
        self._log.propagate = False
        self._sinks: dict[int, logging.Handler] = {}
        self._next_id = 0
        h = logging.StreamHandler(sys.stderr)
        h.setFormatter(logging.Formatter(_FMT))
        h.setLevel(logging.INFO)
        h.addFilter(_RankFilter())
        self._log.addHandler(h)
        self._sinks[self._next_id] = h
        self._next_id += 1

    def add(self, sink: Any, rotation: str | None = None, level: str = "DEBUG", **kw: Any) -> int:
        """loguru-style sink registration. ``rotation`` like "10 MB" is honored."""
        if isinstance(sink, (str, os.PathLike)):
            max_bytes = 0
            if rotation:
                txt = str(rotation).upper().replace(" ", "")
                for suffix, mult in (("MB", 1 << 20), ("KB", 1 << 10), ("GB", 1 << 30), ("B", 1)):
                    if txt.endswith(suffix):
                        max_bytes = int(float(txt[: -len(suffix)]) * mult)
                        break
            h: logging.Handler = logging.handlers.RotatingFileHandler(
                sink, maxBytes=max_bytes, backupCount=3
            )
        else:
            h = logging.StreamHandler(sink)
        h.setFormatter(logging.Formatter(_FMT))
        h.setLevel(getattr(logging, level.upper(), logging.DEBUG))
        h.addFilter(_RankFilter())
        self._log.addHandler(h)
        sid = self._next_id
        self._sinks[sid] = h
        self._next_id += 1
        return sid

    def remove(self, sink_id: int | None = None) -> None:
        if sink_id is None:
            for h in list(self._sinks.values()):
                self._log.removeHandler(h)
            self._sinks.clear()
        elif sink_id in self._sinks:
            self._log.removeHandler(self._sinks.pop(sink_id))

    def debug(self, msg: str, *a: Any) -> None:
        self._log.debug(msg, *a)

    def info(self, msg: str, *a: Any) -> None:
        self._log.info(msg, *a)

    def warning(self, msg: str, *a: Any) -> None:
        self._log.warning(msg, *a)

    def error(self, msg: str, *a: Any) -> None:
        self._log.error(msg, *a)

    def exception(self, msg: str, *a: Any) -> None:
        self._log.exception(msg, *a)


if HAS_LOGURU:
    logger = _loguru_logger  # type: ignore[assignment]
else:
    logger = _ShimLogger()  # type: ignore[assignment]

__all__ = ["logger", "HAS_LOGURU"]
