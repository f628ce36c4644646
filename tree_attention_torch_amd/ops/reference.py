"""Correct eager-mode attention oracle: output + true pre-softmax LSE.

This is the fixed-semantics version of the reference's ``flash_res_lse``
(/root/reference/model.py:60-83). The reference had three defects that this
module repairs:

1. layout bug — it matmuled the (heads, dim) trailing dims because its data
   was laid out (B, T, nh, C); here the canonical layout is ``(B, H, Tq, D)``
   for Q and ``(B, H, Tk, D)`` for K/V, so scores are (B, H, Tq, Tk);
2. LSE was computed on post-softmax probabilities; here it is the true
   row-wise log-sum-exp of the scaled logits — the quantity the tree
   combination mathematically requires (model.py:80 vs. paper Alg. 1);
3. causal masking used ``torch.tril`` on raw scores (a zeroed score still
   contributes e^0); here masked positions get -inf before the softmax.

Everything is computed in fp32 regardless of input dtype: this is the
numerics oracle every HIP kernel (ops/hip/*) is validated against.
"""

from __future__ import annotations

import math

import torch

__all__ = ["flash_res_lse", "attention_reference", "repeat_kv"]


def repeat_kv(x: torch.Tensor, n_rep: int) -> torch.Tensor:
    """Expand (B, Hkv, T, D) KV heads to (B, Hkv*n_rep, T, D) for GQA."""
    if n_rep == 1:
        return x
    b, hkv, t, d = x.shape
    return x[:, :, None, :, :].expand(b, hkv, n_rep, t, d).reshape(b, hkv * n_rep, t, d)


def flash_res_lse(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    softmax_scale: float | None = None,
    is_causal: bool = False,
    q_offset: int = 0,
    kv_offset: int = 0,
) -> tuple[torch.Tensor, torch.Tensor]:
    """Local attention partial: (out, lse).

    Args:
        q: (B, Hq, Tq, D) query block.
        k, v: (B, Hkv, Tk, D) local K/V shard. Hkv may divide Hq (GQA).
        softmax_scale: score scale; default 1/sqrt(D).
        is_causal: mask positions where global key index > global query index.
        q_offset: global position of q[..., 0, :] (for causal masking of a
            query chunk against a KV shard).
        kv_offset: global position of k[..., 0, :] (rank_offset * shard_len).

    Returns:
        out: (B, Hq, Tq, D) fp32 — softmax(QK^T * scale) V over the LOCAL keys.
        lse: (B, Hq, Tq) fp32 — log sum_j exp(scale * q . k_j) over LOCAL keys.
            Rows with no visible key (fully masked) have lse = -inf, out = 0.

    The pair (out, lse) is exactly what the numerically-stable tree combine
    (parallel/combine.py) consumes; combining the per-shard pairs reproduces
    attention over the concatenated keys bit-for-bit in exact arithmetic.
    """
    if softmax_scale is None:
        softmax_scale = 1.0 / math.sqrt(q.shape[-1])
    orig_hq = q.shape[1]
    if k.shape[1] != orig_hq:
        assert orig_hq % k.shape[1] == 0, "Hq must be a multiple of Hkv"
        rep = orig_hq // k.shape[1]
        k = repeat_kv(k, rep)
        v = repeat_kv(v, rep)

    qf = q.float()
    kf = k.float()
    vf = v.float()
    # scores: (B, H, Tq, Tk)
    scores = torch.matmul(qf, kf.transpose(-2, -1)) * softmax_scale
    if is_causal:
        tq, tk = scores.shape[-2], scores.shape[-1]
        qpos = torch.arange(q_offset, q_offset + tq, device=scores.device)
        kpos = torch.arange(kv_offset, kv_offset + tk, device=scores.device)
        mask = kpos[None, :] > qpos[:, None]  # (Tq, Tk) True => masked
        scores = scores.masked_fill(mask, float("-inf"))
    lse = torch.logsumexp(scores, dim=-1)  # (B, H, Tq); -inf for empty rows
    # stable softmax that yields exact zeros for fully-masked rows
    m = scores.amax(dim=-1, keepdim=True)
    m = torch.where(torch.isfinite(m), m, torch.zeros_like(m))
    p = torch.exp(scores - m)
    denom = p.sum(dim=-1, keepdim=True)
    denom = torch.where(denom == 0, torch.ones_like(denom), denom)
    out = torch.matmul(p / denom, vf)
    return out, lse


def attention_reference(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    softmax_scale: float | None = None,
    is_causal: bool = False,
) -> torch.Tensor:
    """Plain full attention (no sharding) — the ground truth for end-to-end
    sharded == unsharded tests."""
    out, _ = flash_res_lse(q, k, v, softmax_scale=softmax_scale, is_causal=is_causal)
    return out
