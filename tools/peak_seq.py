#!/usr/bin/env python3
"""Peak sequence length per node — the second half of BASELINE.json's metric.

Finds the longest KV sequence a node can hold AND decode over (one decode
step must complete), per GPU and summed over ranks. The cap is computed
analytically from free HBM (KV bytes = 2 tensors * Hkv * D * bytes/elem per
token) and then verified by allocating the cache and running a real
tree-attention decode step at that length; a safety margin keeps the probe
below the OOM line so a failed probe backs off instead of killing the box.

Usage (single GPU):
    python tools/peak_seq.py --dtype bf16
    python tools/peak_seq.py --dtype fp8 --kv-heads 4   # GQA 8:1 config 5
Multi-rank (the driver's 8-GPU box):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 tools/peak_seq.py --dtype bf16

Emits ONE JSON line on rank 0:
    {"metric": "peak seq-len per node", "seq_total": ..., "per_gpu": [...],
     "ms_per_decode_step": ..., "config": {...}}

Round-1 measured points this reproduces at N=1: 8.0M tokens bf16 MHA,
32M tokens fp8 GQA8 (profiles/README.md).
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from tree_attention_torch_amd.parallel.pg import cleanup, setup  # noqa: E402
from tree_attention_torch_amd.parallel.tree import tree_attention  # noqa: E402

_BYTES = {"bf16": 2, "fp16": 2, "fp8": 1}
_DT = {"bf16": torch.bfloat16, "fp16": torch.float16,
       "fp8": torch.float8_e4m3fn}


def probe(rank: int, world: int, args) -> None:
    device = torch.device(f"cuda:{int(os.environ.get('LOCAL_RANK', rank))}")
    torch.cuda.set_device(device)
    if world > 1:
        setup(rank, world)
    try:
        hq = args.heads
        hkv = args.kv_heads or hq
        d = args.head_dim
        per_tok = 2 * hkv * d * _BYTES[args.dtype]  # K + V bytes per token

        free, total = torch.cuda.mem_get_info(device)
        # leave headroom for q/out/lse/split-partials/workspace + allocator
        # slack; the decode split-combine partials are S*B*Hq*D fp32 ~ MBs.
        budget = int(free * args.margin)
        t_cap = budget // per_tok
        t_cap = (t_cap // args.align) * args.align

        q_dtype = torch.bfloat16 if args.dtype == "fp8" else _DT[args.dtype]
        q = torch.randn(args.batch, hq, 1, d, dtype=q_dtype, device=device)

        t_local = t_cap
        step_ms = None
        while t_local >= args.align:
            try:
                kv_shape = (args.batch, hkv, t_local, d)
                if args.dtype == "fp8":
                    k = torch.empty(kv_shape, dtype=torch.uint8,
                                    device=device).view(torch.float8_e4m3fn)
                    v = torch.empty(kv_shape, dtype=torch.uint8,
                                    device=device).view(torch.float8_e4m3fn)
                    # fill via bf16 chunks to bound temp memory
                    for lo in range(0, t_local, 1 << 20):
                        hi = min(t_local, lo + (1 << 20))
                        blk = torch.randn(args.batch, hkv, hi - lo, d,
                                          dtype=torch.bfloat16, device=device)
                        k[:, :, lo:hi] = blk.to(torch.float8_e4m3fn)
                        v[:, :, lo:hi] = blk.to(torch.float8_e4m3fn)
                        del blk
                else:
                    k = torch.randn(kv_shape, dtype=_DT[args.dtype],
                                    device=device)
                    v = torch.randn(kv_shape, dtype=_DT[args.dtype],
                                    device=device)
                # one warm + timed decode steps over the full shard
                tree_attention(q, k, v)
                torch.cuda.synchronize()
                if world > 1:
                    torch.distributed.barrier()
                t0 = time.perf_counter()
                for _ in range(args.steps):
                    tree_attention(q, k, v)
                torch.cuda.synchronize()
                if world > 1:
                    torch.distributed.barrier()
                step_ms = (time.perf_counter() - t0) / args.steps * 1e3
                del k, v
                break
            except torch.cuda.OutOfMemoryError:
                k = v = None  # noqa: F841 — drop refs before retrying
                torch.cuda.empty_cache()
                t_local = ((t_local * 15 // 16) // args.align) * args.align
        if step_ms is None:
            raise RuntimeError("could not fit any KV cache")

        t_dev = torch.tensor([t_local], dtype=torch.long, device=device)
        if world > 1:
            gathered = [torch.zeros_like(t_dev) for _ in range(world)]
            torch.distributed.all_gather(gathered, t_dev)
            per_gpu = [int(t.item()) for t in gathered]
        else:
            per_gpu = [t_local]

        if rank == 0:
            print(json.dumps({
                "metric": "peak seq-len per node",
                "seq_total": int(sum(per_gpu)),
                "per_gpu": per_gpu,
                "n_gpus": world,
                "ms_per_decode_step": round(step_ms, 3),
                "unit": "tokens",
                "higher_is_better": True,
                "config": {
                    "batch": args.batch, "heads": hq, "kv_heads": hkv,
                    "head_dim": d, "dtype": args.dtype,
                    "kv_bytes_per_token": per_tok,
                    "margin": args.margin,
                },
            }), flush=True)
    finally:
        cleanup()


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--batch", type=int, default=1)
    p.add_argument("--heads", type=int, default=32)
    p.add_argument("--kv-heads", type=int, default=None)
    p.add_argument("--head-dim", type=int, default=128)
    p.add_argument("--dtype", choices=("bf16", "fp16", "fp8"), default="bf16")
    p.add_argument("--steps", type=int, default=3)
    p.add_argument("--margin", type=float, default=0.94,
                   help="fraction of free HBM given to the KV cache")
    p.add_argument("--align", type=int, default=65536)
    args = p.parse_args()
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    probe(rank, world, args)


if __name__ == "__main__":
    main()
