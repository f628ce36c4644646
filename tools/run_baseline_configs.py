#!/usr/bin/env python3
"""Run the five BASELINE.json configs and emit one JSON line each.

Configs:
  1. B=1 H=2 seq=128 d=64 eager CPU (plumbing check — oracle path)
  2. 1×GPU flash-decode, B=1 H=32 seq=32K d=128 bf16
  3. 8×GPU tree decode, seq 1M sharded (128K/GPU)     [runs at N visible GPUs]
  4. 8×GPU causal prefill seq 256K with overlap        [scaled to N GPUs]
  5. 8×GPU GQA 8:1 + fp8, seq 2M (HBM-cap decode)      [scaled to N GPUs]

Single-process usage covers N=1; for N>1 launch under torch.distributed.run:
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 tools/run_baseline_configs.py
"""

from __future__ import annotations

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from tree_attention_torch_amd.data import make_data  # noqa: E402
from tree_attention_torch_amd.ops.reference import flash_res_lse  # noqa: E402
from tree_attention_torch_amd.parallel.pg import cleanup, setup  # noqa: E402
from tree_attention_torch_amd.parallel.tree import tree_attention  # noqa: E402
from tree_attention_torch_amd.utils.timing import StepTimer  # noqa: E402


def timed(fn, device, steps, warmup):
    for _ in range(warmup):
        fn()
    with StepTimer(device) as t:
        for _ in range(steps):
            fn()
    return t.max_over_ranks() / steps


def emit(rank, name, seconds, tokens, extra):
    if rank == 0:
        print(json.dumps({
            "config": name,
            "ms_per_step": seconds * 1e3,
            "attention_tokens_per_sec": tokens / seconds,
            **extra,
        }), flush=True)


def main():
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    on_gpu = torch.cuda.is_available()
    device = torch.device(f"cuda:{int(os.environ.get('LOCAL_RANK', rank))}") if on_gpu else torch.device("cpu")
    if world > 1:
        setup(rank, world)

    try:
        # --- config 1: CPU eager plumbing ---
        if rank == 0:
            q = torch.randn(1, 2, 1, 64)
            k = torch.randn(1, 2, 128, 64)
            v = torch.randn(1, 2, 128, 64)
            dt = timed(lambda: flash_res_lse(q, k, v), torch.device("cpu"), 20, 3)
            emit(0, "1_cpu_eager_B1H2S128D64", dt, 128, {"device": "cpu"})

        if not on_gpu:
            return

        # --- config 2: 1-GPU flash decode 32K (per-rank; rank 0 reports) ---
        q, k, v = make_data((1, 32, 32768, 128), rank, device, dtype="bf16")
        spin_until = time.time() + 2.0  # DVFS ramp (see bench.py)
        while time.time() < spin_until:
            tree_attention(q, k, v)
        torch.cuda.synchronize()
        dt = timed(lambda: tree_attention(q[:, :, :1].contiguous(), k, v)
                   if False else tree_attention(q, k, v), device, 50, 10)
        emit(rank, "2_decode_32k_bf16_local", dt, 32768 * world,
             {"n_gpus": world, "dtype": "bf16"})

        # --- config 3: tree decode, 128K KV per GPU (1M at 8 GPUs) ---
        q, k, v = make_data((1, 32, 131072, 128), rank, device, dtype="bf16")
        dt = timed(lambda: tree_attention(q, k, v), device, 30, 5)
        emit(rank, "3_tree_decode_128kpergpu_bf16", dt, 131072 * world,
             {"n_gpus": world, "seq_total": 131072 * world, "dtype": "bf16"})

        # --- config 4: causal prefill, 32K queries/GPU (256K at 8 GPUs),
        #     chunked with compute/collective overlap ---
        t_local = 32768
        q, k, v = make_data((1, 32, t_local, 128), rank, device,
                            q_len=t_local, dtype="bf16")
        dt = timed(lambda: tree_attention(q, k, v, is_causal=True,
                                          overlap=True),
                   device, 5, 2)
        emit(rank, "4_causal_prefill_32kpergpu_bf16", dt, t_local * world,
             {"n_gpus": world, "seq_total": t_local * world, "dtype": "bf16",
              "note": "queries+keys sharded-equivalent: full causal prefill"})

        # --- config 5: GQA 8:1 + fp8 KV, 2M tokens total ---
        t_local = 2097152 // world
        q, k, v = make_data((1, 32, t_local, 128), rank, device,
                            dtype="fp8", kv_heads=4)
        dt = timed(lambda: tree_attention(q, k, v), device, 20, 5)
        emit(rank, "5_decode_2M_gqa8_fp8", dt, t_local * world,
             {"n_gpus": world, "seq_total": t_local * world, "dtype": "fp8",
              "kv_heads": 4})
    finally:
        cleanup()


if __name__ == "__main__":
    main()
