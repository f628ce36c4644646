"""Driver: per-GPU spawn, data generation, timed tree decode.

Reference parity for ``main`` + the ``__main__`` entry
(/root/reference/model.py:129-169): spawn one process per GPU
(torch.multiprocessing), loguru-style file sink with 10 MB rotation, CPU
single-process fallback — with the measurement fixed (device-synchronized
timing) and the workload configurable (config.py).

Run:  python -m tree_attention_torch_amd [--seq-len N] [--num-heads H] ...
"""

from __future__ import annotations

import sys

import torch
import torch.multiprocessing as mp

from .config import TreeAttentionConfig
from .data import make_data
from .parallel.pg import cleanup, local_device, setup
from .parallel.tree import tree_attention
from .utils.logging import logger
from .utils.timing import StepTimer


def main(rank: int, world_size: int, cfg: TreeAttentionConfig | None = None) -> None:
    """Run one rank's share of a timed tree-attention decode."""
    cfg = cfg or TreeAttentionConfig()
    device = local_device(rank)
    addr, port = cfg.env_master()
    setup(rank, world_size, backend=cfg.backend, master_addr=addr, master_port=port)
    try:
        t_local = cfg.seq_len // max(world_size, 1)
        q, k, v = make_data(
            (cfg.batch, cfg.num_heads, t_local, cfg.head_dim),
            rank,
            device,
            q_len=cfg.q_len,
            dtype=cfg.dtype,
            kv_heads=cfg.kv_heads,
            seed=cfg.seed,
        )
        logger.info(
            f"Rank {rank}: starting seq_len={cfg.seq_len} (local {t_local}) "
            f"H={cfg.num_heads} D={cfg.head_dim} dtype={cfg.dtype}"
        )
        # warmup (kernel compilation, RCCL communicator setup)
        for _ in range(cfg.warmup):
            tree_attention(q, k, v, softmax_scale=cfg.scale, is_causal=cfg.causal,
                           combine=cfg.combine)
        with StepTimer(device) as t:
            for _ in range(cfg.steps):
                out = tree_attention(q, k, v, softmax_scale=cfg.scale,
                                     is_causal=cfg.causal, combine=cfg.combine)
        elapsed = t.max_over_ranks()
        per_step = elapsed / cfg.steps
        tok_s = cfg.batch * cfg.seq_len * cfg.steps / elapsed
        logger.info(
            f"Rank {rank}: {cfg.steps} steps in {elapsed:.4f}s "
            f"({per_step * 1e3:.3f} ms/step, {tok_s:.3e} attention tokens/s), "
            f"out shape {tuple(out.shape)}"
        )
    finally:
        cleanup()


def entry(argv: list[str] | None = None) -> None:
    cfg = TreeAttentionConfig.from_args(argv)
    logger.add("tree_attention_log.log", rotation="10 MB")
    if torch.cuda.is_available():
        world_size = torch.cuda.device_count()
        logger.info(f"Running on {world_size} GPUs.")
        if world_size > 1:
            mp.spawn(main, args=(world_size, cfg), nprocs=world_size, join=True)
        else:
            main(0, 1, cfg)
    else:
        logger.info("Running on CPU.")
        main(0, 1, cfg)


if __name__ == "__main__":
    entry(sys.argv[1:])
