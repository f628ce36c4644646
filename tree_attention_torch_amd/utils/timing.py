"""Device-correct timing helpers.

The reference timed ``tree_decode`` with bare ``time.time()`` and no device
synchronization (/root/reference/model.py:149-153) — on GPU that measures
launch latency, not kernel time. These helpers bracket
timed regions with a barrier + full device sync on both sides and report the
MAX over ranks, which is what bench.py's contract requires.
"""

from __future__ import annotations

import time

import torch
import torch.distributed as dist


def barrier_sync(device: torch.device | None = None) -> None:
    sync = torch.cuda.is_available() and (device is None or device.type == "cuda")
    if sync:
        torch.cuda.synchronize(device)
    if dist.is_available() and dist.is_initialized():
        dist.barrier()
    if sync:
        torch.cuda.synchronize(device)


class StepTimer:
    """Wall-clock a region with proper sync; max-reduced across ranks."""

    def __init__(self, device: torch.device | None = None):
        self.device = device
        self.t0 = 0.0
        self.elapsed = 0.0

    def __enter__(self) -> "StepTimer":
        barrier_sync(self.device)
        self.t0 = time.perf_counter()
        return self

    def __exit__(self, *exc) -> None:
        barrier_sync(self.device)
        self.elapsed = time.perf_counter() - self.t0

    def max_over_ranks(self) -> float:
        if dist.is_available() and dist.is_initialized():
            # NCCL/RCCL reduces CUDA tensors only; gloo wants CPU
            dev = "cuda" if (torch.cuda.is_available()
                             and dist.get_backend() == "nccl") else "cpu"
            t = torch.tensor([self.elapsed], dtype=torch.float64, device=dev)
            dist.all_reduce(t, op=dist.ReduceOp.MAX)
            return float(t.item())
        return self.elapsed


__all__ = ["StepTimer", "barrier_sync"]
