"""Synthetic Q/K/V generation, per-rank sharded.

Mirrors the reference's ``make_data`` (/root/reference/model.py:37-56) with
the intended semantics: Q is REPLICATED across ranks (the paper shards only
K/V; the reference's seed-by-rank also perturbed Q — SURVEY.md §2.3), K/V are
per-rank shards realized by a per-rank seed, and the layout is the canonical
``(B, H, T, D)`` the reference documented but did not produce (model.py:42).
"""

from __future__ import annotations

import torch

_DTYPES = {
    "fp16": torch.float16,
    "bf16": torch.bfloat16,
    "fp32": torch.float32,
}


def make_data(
    shape: tuple[int, int, int, int],
    rank: int,
    device: torch.device | str,
    q_len: int = 1,
    dtype: str | torch.dtype = "bf16",
    kv_heads: int | None = None,
    seed: int = 0,
) -> tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Generate (Q, K, V) for one rank.

    Args:
        shape: (B, H, T_local, D) — T_local is THIS RANK's KV shard length.
        rank: distributed rank; K/V use seed ``seed + 1 + rank`` so each rank
            draws a distinct shard (the per-rank seed IS the sharding
            mechanism, as in the reference model.py:50); Q uses ``seed`` on
            every rank so it is replicated.
        q_len: number of query positions (1 = decode).
        dtype: storage dtype for Q/K/V.
        kv_heads: number of KV heads (GQA); None => H.

    Returns:
        Q (B, H, q_len, D), K (B, Hkv, T_local, D), V (B, Hkv, T_local, D)
        on ``device``.
    """
    b, h, t, d = shape
    hkv = kv_heads if kv_heads is not None else h
    if isinstance(dtype, str) and dtype == "fp8":
        # fp8 KV cache (OCP e4m3), bf16 queries — BASELINE config 5
        q_td, td = torch.bfloat16, torch.float8_e4m3fn
    else:
        q_td = td = _DTYPES[dtype] if isinstance(dtype, str) else dtype
    dev = torch.device(device)
    # generate directly on the target device: at T=128K per shard the K+V
    # payload is ~2 GB bf16 — a host round-trip (reference model.py:51-53)
    # would serialize ranks behind PCIe for no reason.
    gen_dev = dev if dev.type == "cuda" else torch.device("cpu")
    g = torch.Generator(device=gen_dev)
    # draw directly in a 2-byte dtype (fp8 can't be drawn: go through bf16):
    # an fp32 staging draw would need 2x the KV bytes — at HBM-cap sequences
    # (tens of GB of KV per GPU) that OOMs long before the cache does.
    draw_td = td if td in (torch.bfloat16, torch.float16, torch.float32) \
        else torch.bfloat16
    g.manual_seed(seed)
    q = torch.randn((b, h, q_len, d), generator=g, device=gen_dev, dtype=q_td)
    g.manual_seed(seed + 1 + rank)
    k = torch.randn((b, hkv, t, d), generator=g, device=gen_dev, dtype=draw_td)
    v = torch.randn((b, hkv, t, d), generator=g, device=gen_dev, dtype=draw_td)
    if k.dtype != td:
        k = k.to(td)
        v = v.to(td)
    return (q.to(dev), k.to(dev), v.to(dev))


__all__ = ["make_data"]
