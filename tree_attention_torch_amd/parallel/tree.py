"""Tree attention: sequence-sharded attention with log-sum-exp tree combine.

The distributed algorithm layer (reference ``tree_decode``,
/root/reference/model.py:85-124 — whose multi-GPU branch never ran,
SURVEY.md §0.1.4). Per rank r holding KV shard (K_r, V_r) and replicated Q:

    (out_r, lse_r) = local_attention(Q, K_r, V_r)     # HIP kernel on GPU
    out            = tree_combine(out_r, lse_r)       # RCCL over xGMI

Decode (Tq == 1) is one kernel + one latency-bound collective. Causal
prefill chunks Q and overlaps chunk i's collective with chunk i+1's kernel
(SURVEY.md §5.7): compute runs on the current stream while the combine of
the previous chunk proceeds; RCCL internally uses its own streams so the
async handle is the ordering point.
"""

from __future__ import annotations

import math

import torch
import torch.distributed as dist

from ..ops.flash import local_attention
from .combine import tree_combine

__all__ = ["tree_decode", "tree_attention", "TreeAttention"]


def default_q_chunk(tq: int, b: int, hq: int) -> int:
    """Prefill chunk size (pure; pinned by tests/test_routing.py).

    Keep every chunk's grid full: the prefill kernel launches
    B*Hq*(chunk/256) blocks and the chip wants >= 512 (a 4096-row chunk at
    B*Hq=8 is 128 blocks — measured 5.5x slower than unchunked). When
    chunks DO fill the grid, chunking is mildly faster even at world 1
    (causal-tail scheduling: 13.4 vs 14.2 ms at H=32, 32K rows, same box)
    and at world > 1 it is what overlaps each chunk's collective with the
    next kernel.
    """
    min_chunk = max(4096, (512 * 256) // max(b * hq, 1))
    return tq if tq <= min_chunk else min_chunk


def _rank_and_world(group) -> tuple[int, int]:
    if dist.is_available() and dist.is_initialized():
        return dist.get_rank(group), dist.get_world_size(group)
    return 0, 1


def tree_attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    softmax_scale: float | None = None,
    is_causal: bool = False,
    combine: str = "auto",
    group: dist.ProcessGroup | None = None,
    q_chunk: int | None = None,
    overlap: bool = True,
    return_lse: bool = False,
    kv_offset: int | None = None,
    total_kv: int | None = None,
    kv_scales: tuple[torch.Tensor, torch.Tensor] | None = None,
):
    """Attention over the KV sequence sharded across ranks.

    Args:
        q: (B, Hq, Tq, D) — replicated across ranks.
        k, v: (B, Hkv, T_local, D) — THIS rank's shard; rank r holds global
            positions [r*T_local, (r+1)*T_local).
        kv_scales: optional (k_scales, v_scales) for an MX block-scaled
            fp8 shard (quant.quantize_k_mx / quantize_v_mx) — the local
            partial runs the hardware-scaled kernel on GPU and the
            dequantized fp32 oracle on CPU; the combine is unchanged
            (lse algebra is dtype-agnostic).
        is_causal: causal masking in GLOBAL positions; queries are taken to
            be the LAST Tq positions of the global sequence (decode
            semantics: a new token attends to everything before it).
        combine: "auto" | "allgather" | "allreduce" (parallel/combine.py).
        q_chunk: prefill chunk size along Tq (None = pick automatically).
        overlap: overlap chunk i's collective with chunk i+1's kernel.
        kv_offset / total_kv: global position of this rank's shard and the
            global sequence length. Default (None) assumes EQUAL shards —
            rank r holds [r*T_local, (r+1)*T_local). Pass both explicitly
            for uneven sharding (e.g. ragged serving caches); the combine
            is shard-size-agnostic, only the causal positions need them.

    Returns:
        out (B, Hq, Tq, D) fp32 (and lse (B, Hq, Tq) if return_lse).
    """
    rank, world = _rank_and_world(group)
    t_local = k.shape[-2]
    if (kv_offset is None) != (total_kv is None):
        raise ValueError("pass kv_offset and total_kv together (or neither)")
    if kv_offset is None:
        kv_offset = rank * t_local
        total_kv = world * t_local
    tq = q.shape[-2]
    if tq > total_kv:
        raise ValueError(
            f"tree_attention: q_len {tq} exceeds the global KV length "
            f"{total_kv} — queries are the LAST Tq positions of the global "
            "sequence (decode semantics), so each query needs a KV position"
        )
    # queries sit at the END of the global sequence (prefill over the full
    # sequence has tq == total_kv and q_offset 0).
    q_offset = total_kv - tq

    if softmax_scale is None:
        softmax_scale = 1.0 / math.sqrt(q.shape[-1])

    if q_chunk is None:
        q_chunk = default_q_chunk(tq, q.shape[0], q.shape[1])

    def _local(qc, q_off_c):
        if kv_scales is None:
            return local_attention(qc, k, v, softmax_scale, is_causal,
                                   q_off_c, kv_offset)
        ks, vs = kv_scales
        if qc.device.type == "cuda":
            from ..ops.flash import local_attention_mx

            return local_attention_mx(qc, k, ks, v, vs, softmax_scale,
                                      is_causal, q_off_c, kv_offset)
        from ..ops.reference import flash_res_lse as oracle
        from ..quant import dequantize_k_mx, dequantize_v_mx

        return oracle(qc.float(), dequantize_k_mx(k, ks),
                      dequantize_v_mx(v, vs), softmax_scale, is_causal,
                      q_off_c, kv_offset)

    if tq <= q_chunk:
        out_l, lse_l = _local(q, q_offset)
        out, lse = tree_combine(out_l, lse_l, strategy=combine, group=group)
        return (out, lse) if return_lse else out

    # chunked prefill with compute/collective overlap
    outs: list[torch.Tensor | None] = []
    lses: list[torch.Tensor | None] = []
    pending = None  # (index, handle)
    n_chunks = (tq + q_chunk - 1) // q_chunk
    for c in range(n_chunks):
        lo = c * q_chunk
        hi = min(tq, lo + q_chunk)
        out_c, lse_c = _local(q[..., lo:hi, :], q_offset + lo)
        outs.append(None)
        lses.append(None)
        if overlap:
            if pending is not None:
                i, h = pending
                outs[i], lses[i] = h.wait()
            pending = (c, tree_combine(out_c, lse_c, strategy=combine, group=group, async_op=True))
        else:
            outs[c], lses[c] = tree_combine(out_c, lse_c, strategy=combine, group=group)
    if pending is not None:
        i, h = pending
        outs[i], lses[i] = h.wait()
    out = torch.cat(outs, dim=-2)
    lse = torch.cat(lses, dim=-2)
    return (out, lse) if return_lse else out


def tree_decode(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    rank: int | None = None,
    world_size: int | None = None,
    device: torch.device | None = None,
    softmax_scale: float | None = None,
    combine: str = "auto",
) -> torch.Tensor:
    """Reference-compatible entry (model.py:85): single-query decode over the
    sharded KV. rank/world_size/device are accepted for API parity but are
    derived from the process group when omitted."""
    del rank, world_size, device  # derived from the process group
    return tree_attention(q, k, v, softmax_scale=softmax_scale, combine=combine)


class TreeAttention(torch.nn.Module):
    """Module wrapper: holds scale/causal/combine config; forward = tree_attention."""

    def __init__(
        self,
        softmax_scale: float | None = None,
        causal: bool = False,
        combine: str = "auto",
        q_chunk: int | None = None,
        overlap: bool = True,
    ) -> None:
        super().__init__()
        self.softmax_scale = softmax_scale
        self.causal = causal
        self.combine = combine
        self.q_chunk = q_chunk
        self.overlap = overlap

    def forward(self, q: torch.Tensor, k: torch.Tensor, v: torch.Tensor) -> torch.Tensor:
        return tree_attention(
            q,
            k,
            v,
            softmax_scale=self.softmax_scale,
            is_causal=self.causal,
            combine=self.combine,
            q_chunk=self.q_chunk,
            overlap=self.overlap,
        )
