"""Process-group lifecycle: one process per GPU over RCCL/xGMI.

Reference surface: ``setup``/``cleanup`` (/root/reference/model.py:11-33).
Differences by design:

* CPU multi-process is REAL here (gloo backend) — the reference silently
  skipped distribution on CPU (model.py:19), so its combine math was never
  exercised; our gloo path is the "multi-node without a cluster" test
  mechanism.
* The backend string "nccl" on ROCm *is* RCCL; rendezvous defaults to
  127.0.0.1 and honors MASTER_ADDR/MASTER_PORT (torchrun compatible).
* Failure handling: a timeout is configured so a dead rank aborts the
  collective instead of hanging forever, and ``cleanup`` is safe to call on
  the error path.
"""

from __future__ import annotations

import datetime
import os

import torch
import torch.distributed as dist

from ..utils.logging import logger

__all__ = ["setup", "cleanup", "is_distributed", "local_device"]


def setup(
    rank: int,
    world_size: int,
    backend: str | None = None,
    master_addr: str = "127.0.0.1",
    master_port: int = 12355,
    timeout_s: float = 600.0,
    device: torch.device | None = None,
) -> None:
    """Initialize the process group for this rank.

    On GPU: backend "nccl" (= RCCL over xGMI on ROCm), device binding to
    cuda:{LOCAL_RANK or rank}. On CPU: gloo. No-op when world_size == 1 and
    no env rendezvous is configured, matching the reference's single-process
    fallback (model.py:166-169).
    """
    if world_size <= 1 and "MASTER_ADDR" not in os.environ:
        logger.info("setup: world_size=1, running without a process group")
        return
    if dist.is_initialized():
        return
    os.environ.setdefault("MASTER_ADDR", master_addr)
    os.environ.setdefault("MASTER_PORT", str(master_port))
    if backend is None:
        # TREE_ATTN_BACKEND=gloo lets multi-rank GPU code paths run with
        # several ranks sharing one device (RCCL refuses duplicate devices)
        backend = os.environ.get(
            "TREE_ATTN_BACKEND",
            "nccl" if torch.cuda.is_available() else "gloo",
        )
    kwargs = {}
    if backend == "nccl":
        local = int(os.environ.get("LOCAL_RANK", rank))
        torch.cuda.set_device(local)
        kwargs["device_id"] = torch.device(f"cuda:{local}")
    elif backend == "gloo" and torch.cuda.is_available():
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", 0)))
    dist.init_process_group(
        backend,
        rank=rank,
        world_size=world_size,
        timeout=datetime.timedelta(seconds=timeout_s),
        **kwargs,
    )
    logger.info(
        f"Distributed environment initialized: backend={backend} "
        f"rank={rank}/{world_size}"
    )


def cleanup() -> None:
    """Destroy the process group; safe to call when uninitialized or on the
    error path (reference: model.py:27-33)."""
    if dist.is_available() and dist.is_initialized():
        dist.destroy_process_group()
        logger.info("Distributed environment cleaned up.")


def is_distributed() -> bool:
    return dist.is_available() and dist.is_initialized() and dist.get_world_size() > 1


def local_device(rank: int | None = None) -> torch.device:
    """cuda:{LOCAL_RANK} when GPUs are visible, else CPU (model.py:137)."""
    if torch.cuda.is_available():
        local = int(os.environ.get("LOCAL_RANK", rank if rank is not None else 0))
        return torch.device(f"cuda:{local}")
    return torch.device("cpu")
