#!/usr/bin/env python3
"""Flagship benchmark: tree-attention decode over a sharded KV sequence.

Contract (driver-facing):
    python bench.py --gpus N --steps K --warmup W
For N > 1 the driver launches this file under torch.distributed.run with one
rank per GPU over RCCL; rendezvous comes from RANK/LOCAL_RANK/WORLD_SIZE/
MASTER_* env vars. W untimed warmup steps, then EXACTLY K timed steps
bracketed by barrier + torch.cuda.synchronize on both sides; elapsed is the
MAX over ranks; rank 0 prints ONE JSON line.

Metric (BASELINE.json): attention tokens/sec at B=1, H=32, d=128, bf16 —
the decode step attends one query over the whole sharded KV sequence, so
value = B * seq_total * steps / elapsed = KV tokens attended per second,
aggregated over the whole job. Weak scaling: each GPU holds a fixed
128K-token KV shard (BASELINE.json config 3: 8 GPUs => seq 1M).
Synthetic random Q/K/V (no datasets exist offline); no weights (the
reference computes attention over raw tensors — SURVEY.md §2.3).
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from tree_attention_torch_amd.data import make_data  # noqa: E402
from tree_attention_torch_amd.parallel.pg import cleanup, setup  # noqa: E402
from tree_attention_torch_amd.parallel.tree import tree_attention  # noqa: E402
from tree_attention_torch_amd.utils.timing import StepTimer  # noqa: E402


def parse_args(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=50)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--seq-per-gpu", type=int, default=131072,
                   help="KV shard length per GPU (weak scaling)")
    p.add_argument("--batch", type=int, default=1)
    p.add_argument("--heads", type=int, default=32)
    p.add_argument("--kv-heads", type=int, default=None)
    p.add_argument("--head-dim", type=int, default=128)
    p.add_argument("--dtype", type=str, default="bf16")
    p.add_argument("--q-len", type=int, default=1)
    p.add_argument("--causal", action="store_true")
    p.add_argument("--combine", type=str, default="auto")
    p.add_argument("--profile", type=str, default=None, metavar="DIR",
                   help="export a torch.profiler chrome trace of a few steps")
    return p.parse_args(argv)


def run(rank: int, world: int, args) -> None:
    on_gpu = torch.cuda.is_available()
    if not on_gpu and args.seq_per_gpu > 8192:
        args.seq_per_gpu = 2048  # CPU smoke: keep it minutes, not hours
    device = torch.device(f"cuda:{int(os.environ.get('LOCAL_RANK', rank))}") if on_gpu \
        else torch.device("cpu")
    if world > 1:
        setup(rank, world)
    try:
        q, k, v = make_data(
            (args.batch, args.heads, args.seq_per_gpu, args.head_dim),
            rank,
            device,
            q_len=args.q_len,
            dtype=args.dtype,
            kv_heads=args.kv_heads,
        )

        def step():
            return tree_attention(q, k, v, is_causal=args.causal, combine=args.combine)

        for _ in range(args.warmup):
            step()
        if on_gpu:
            # DVFS ramp: a few warmup steps (~ms of activity) leave the GPU
            # at idle clocks and the first timed steps ~15% slow (measured:
            # cold bench 0.349 ms/step vs 0.318 after a prior run on the
            # same box). Spin the SAME step untimed for ~2 s wall clock.
            # Synchronize every batch of spins so perf_counter tracks DEVICE
            # time (async launches would otherwise fill the queue and end
            # the "2 s" ramp after ~0.2 s of real GPU activity — the round-1
            # cold-box 0.318-0.358 ms/step spread came from that).
            spin_until = time.perf_counter() + 2.0
            while time.perf_counter() < spin_until:
                for _ in range(50):
                    step()
                torch.cuda.synchronize()
        if args.profile and rank == 0:
            # tracing subsystem: kernel-level chrome trace
            from torch.profiler import ProfilerActivity, profile

            with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA]) as prof:
                for _ in range(3):
                    step()
                torch.cuda.synchronize()
            os.makedirs(args.profile, exist_ok=True)
            prof.export_chrome_trace(os.path.join(args.profile, "bench_trace.json"))
        with StepTimer(device) as t:
            for _ in range(args.steps):
                step()
        elapsed = t.max_over_ranks()

        if rank == 0:
            seq_total = args.seq_per_gpu * world
            tokens = args.batch * seq_total * args.steps
            value = tokens / elapsed
            kv_bytes_per_tok = ((args.kv_heads or args.heads) * args.head_dim
                                * 2 * (1 if args.dtype == "fp8" else 2))
            eff_tbps = value * kv_bytes_per_tok / 1e12
            result = {
                "metric": "attention tokens/sec",
                "value": value,
                "unit": "tokens/s",
                "n_gpus": world,
                "steps": args.steps,
                "warmup": args.warmup,
                "ms_per_step": elapsed / args.steps * 1e3,
                "higher_is_better": True,
                "scaling": "weak",
                "vs_baseline": None,
                "dtype": args.dtype,
                "data": "synthetic",
                "config": {
                    "model": "tree-attention-decode",
                    "global_batch": args.batch,
                    "seq_len": seq_total,
                    "seq_per_gpu": args.seq_per_gpu,
                    "num_heads": args.heads,
                    "kv_heads": args.kv_heads or args.heads,
                    "head_dim": args.head_dim,
                    "q_len": args.q_len,
                    "parallelism": f"sp{world}",
                    "combine": args.combine,
                    "effective_kv_tbps_per_gpu": round(eff_tbps / world, 3),
                    "device": "MI355X" if on_gpu else "cpu",
                },
            }
            print(json.dumps(result), flush=True)
    finally:
        cleanup()


def _spawn_worker(rank: int, world: int, args, port: int):
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    run(rank, world, args)


def main(argv=None) -> None:
    args = parse_args(argv)
    env_world = int(os.environ.get("WORLD_SIZE", "0"))
    if env_world > 1:
        # launched under torch.distributed.run: one process per GPU already
        rank = int(os.environ.get("RANK", "0"))
        run(rank, env_world, args)
    elif args.gpus > 1:
        # standalone multi-GPU convenience: spawn ourselves
        import torch.multiprocessing as mp

        port = 29000 + int(time.time()) % 1000
        mp.spawn(_spawn_worker, args=(args.gpus, args, port), nprocs=args.gpus, join=True)
    else:
        run(0, 1, args)


if __name__ == "__main__":
    main()
