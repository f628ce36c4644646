"""A/B: chunked-prefill compute/collective overlap.

Two ranks share one GPU over gloo (CUDA tensors; collectives bounce via
CPU — RCCL refuses duplicate devices). This makes the collective SLOW and
the overlap benefit VISIBLE if the async-handle path really lets chunk
k+1's kernels run under chunk k's combine: with overlap off, wall ~=
sum(compute) + sum(combine); with overlap on, wall ~= sum(compute) +
one combine + max(0, combine - compute) overlaps. A gloo-GPU proxy —
the absolute numbers are not RCCL numbers, but the DELTA is the overlap
machinery working (handle issued before the next local_attention, waited
one chunk late). The RCCL version of this A/B runs in
tests/test_gpu_multirank.py on any >=2-GPU box.
"""
import os
import sys
import time

import torch
import torch.multiprocessing as mp


def worker(rank, q):
    sys.path.insert(0, "/root/repo")
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT="29761",
                      RANK=str(rank), WORLD_SIZE="2", LOCAL_RANK="0",
                      TREE_ATTN_BACKEND="gloo")
    from tree_attention_torch_amd.parallel.pg import cleanup, setup
    from tree_attention_torch_amd.parallel.tree import tree_attention

    setup(rank, 2)
    try:
        torch.manual_seed(3)
        tl, tq, chunk = 32768, 16384, 4096
        qq = torch.randn(1, 32, tq, 128).bfloat16().cuda()
        k = torch.randn(1, 32, tl, 128).bfloat16().cuda()
        v = torch.randn(1, 32, tl, 128).bfloat16().cuda()
        res = {}
        for overlap in (False, True):
            for _ in range(2):
                tree_attention(qq, k, v, is_causal=True, q_chunk=chunk,
                               overlap=overlap)
            torch.cuda.synchronize()
            torch.distributed.barrier()
            t0 = time.perf_counter()
            for _ in range(4):
                tree_attention(qq, k, v, is_causal=True, q_chunk=chunk,
                               overlap=overlap)
            torch.cuda.synchronize()
            torch.distributed.barrier()
            res[overlap] = (time.perf_counter() - t0) / 4 * 1e3
        if rank == 0:
            gain = (res[False] - res[True]) / res[False] * 100
            print(f"overlap A/B (gloo-GPU ws=2, tq={tq} chunk={chunk} "
                  f"tl={tl}): off={res[False]:.1f} ms on={res[True]:.1f} ms "
                  f"({gain:+.1f}%)", flush=True)
    finally:
        cleanup()


if __name__ == "__main__":
    mp.spawn(worker, args=(None,), nprocs=2, join=True)
