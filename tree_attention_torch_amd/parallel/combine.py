"""Numerically-stable combination of attention partials (out, lse).

The algorithmic core of tree attention: for shard partials
(out_r, lse_r) over disjoint key sets, attention over the union is

    m   = max_r lse_r
    out = sum_r out_r * exp(lse_r - m) / sum_r exp(lse_r - m)
    lse = m + log sum_r exp(lse_r - m)

which is associative and commutative — reducible in any tree order, which is
why an all-reduce (whose implementation IS a topology-aware tree/ring) can
realize it. The reference expressed this with three serialized NCCL
all-reduces on wastefully-expanded buffers and a shape bug that crashed the
multi-GPU path (/root/reference/model.py:105-116). Here:

* ``combine_partials``     — local reduction over a stacked split dimension
                             (used by the split-KV decode kernel's epilogue
                             and as the oracle for the collective paths);
* ``tree_combine_allreduce`` — MAX all-reduce on lse + ONE packed SUM
                             all-reduce (num ‖ den fused into one buffer:
                             2 collectives, not the reference's 3);
* ``tree_combine_allgather`` — ONE all-gather of the packed (out, lse) pair
                             + local combine. For decode-sized payloads
                             (~KBs) collective latency dominates, so the
                             single-collective form wins;
* ``tree_combine``          — strategy dispatch ("auto" picks by payload).

All combine math runs in fp32 regardless of input dtype (SURVEY.md §7 hard
part 3: the (lse, max) carry stays fp32 end-to-end).
"""

from __future__ import annotations

import torch
import torch.distributed as dist

__all__ = [
    "combine_partials",
    "tree_combine",
    "tree_combine_allreduce",
    "tree_combine_allgather",
]

# below ~exp(-80) a shard's contribution underflows fp32 relative to the max;
# clamping keeps exp() finite when a fully-masked shard reports lse = -inf.
_NEG_CLAMP = -80.0


def auto_strategy(out_numel: int) -> str:
    """Combine-strategy pick (pure; pinned by tests/test_routing.py):
    allgather when the fp32 payload is latency-bound (< 1 MiB, decode
    sizes), allreduce when bandwidth-bound (prefill chunks)."""
    return "allgather" if out_numel * 4 < (1 << 20) else "allreduce"


def _rescale(out: torch.Tensor, lse: torch.Tensor, m: torch.Tensor):
    """num = out * w, den = w with w = exp(lse - m), safe for lse = m = -inf."""
    w = torch.exp(torch.clamp(lse - m, min=_NEG_CLAMP, max=0.0))
    w = torch.where(torch.isfinite(lse), w, torch.zeros_like(w))
    return out * w.unsqueeze(-1), w


def combine_partials(
    outs: torch.Tensor, lses: torch.Tensor
) -> tuple[torch.Tensor, torch.Tensor]:
    """Reduce S split partials locally.

    Args:
        outs: (S, B, H, Tq, D) fp32 partial outputs.
        lses: (S, B, H, Tq) fp32 partial LSEs.
    Returns:
        (out, lse): (B, H, Tq, D), (B, H, Tq) — attention over the union.
    """
    outs = outs.float()
    lses = lses.float()
    m = lses.amax(dim=0)  # (B, H, Tq)
    m_safe = torch.where(torch.isfinite(m), m, torch.zeros_like(m))
    w = torch.exp(torch.clamp(lses - m_safe.unsqueeze(0), min=_NEG_CLAMP))
    w = torch.where(torch.isfinite(lses), w, torch.zeros_like(w))
    den = w.sum(dim=0)  # (B, H, Tq)
    num = (outs * w.unsqueeze(-1)).sum(dim=0)  # (B, H, Tq, D)
    den_safe = torch.where(den == 0, torch.ones_like(den), den)
    out = num / den_safe.unsqueeze(-1)
    lse = m_safe + torch.log(den_safe)
    lse = torch.where(den == 0, torch.full_like(lse, float("-inf")), lse)
    return out, lse


def tree_combine_allreduce(
    out: torch.Tensor,
    lse: torch.Tensor,
    group: dist.ProcessGroup | None = None,
    async_op: bool = False,
):
    """MAX all-reduce on lse, then ONE packed SUM all-reduce on [num ‖ den].

    Equivalent of the reference's model.py:105-116 with the intended compact
    shapes ((B,H,Tq) lse, not lse expanded to the output shape) and the N3+N4
    collectives fused into one (SURVEY.md §2.2 design note).

    Returns (out, lse) if async_op is False, else a handle with .wait() ->
    (out, lse) so prefill can overlap the collective with the next chunk's
    compute.
    """
    out = out.float().contiguous()
    lse = lse.float().contiguous()
    m = torch.where(torch.isfinite(lse), lse, torch.full_like(lse, _NEG_CLAMP))
    dist.all_reduce(m, op=dist.ReduceOp.MAX, group=group)
    b, h, tq, d = out.shape
    ext = None
    if out.is_cuda:
        from ..ops import flash

        ext = flash._load_extension()
    if ext is not None:
        # one fused pass over the payload instead of the eager
        # exp/clamp/where/mul/cat chain (67 MB per 4096-row prefill chunk)
        packed = ext.combine_rescale_pack(out, lse.contiguous(), m.contiguous())
    else:
        num, den = _rescale(out, lse, m)
        packed = torch.cat([num.reshape(b, h, tq, d), den.unsqueeze(-1)], dim=-1)
        packed = packed.contiguous()
    work = dist.all_reduce(packed, op=dist.ReduceOp.SUM, group=group, async_op=async_op)

    def _finish():
        if ext is not None:
            return ext.combine_finish(packed, m.contiguous())
        num_g = packed[..., :d]
        den_g = packed[..., d]
        den_safe = torch.where(den_g == 0, torch.ones_like(den_g), den_g)
        out_g = num_g / den_safe.unsqueeze(-1)
        lse_g = m + torch.log(den_safe)
        lse_g = torch.where(den_g == 0, torch.full_like(lse_g, float("-inf")), lse_g)
        return out_g, lse_g

    if async_op:
        class _Handle:
            def wait(self):
                work.wait()
                return _finish()

        return _Handle()
    return _finish()


def tree_combine_allgather(
    out: torch.Tensor,
    lse: torch.Tensor,
    group: dist.ProcessGroup | None = None,
    async_op: bool = False,
):
    """ONE all-gather of the packed (out, lse) pair + local stable combine.

    For 8 ranks at decode sizes (B=1 H=32 D=128 => ~16.5 KB fp32 per rank)
    a single latency-bound collective beats two serialized ones; the local
    combine over 8 gathered partials is trivial.
    """
    out = out.float().contiguous()
    lse = lse.float().contiguous()
    b, h, tq, d = out.shape
    packed = torch.cat([out, lse.unsqueeze(-1)], dim=-1).contiguous()  # (B,H,Tq,D+1)
    ws = dist.get_world_size(group)
    flat = packed.reshape(-1)
    gathered = torch.empty(ws * flat.numel(), dtype=packed.dtype, device=packed.device)
    work = dist.all_gather_into_tensor(gathered, flat, group=group, async_op=async_op)

    def _finish():
        if gathered.is_cuda and d in (64, 128):
            # one HIP kernel instead of the eager amax/exp/sum/div/log
            # chain — the per-step epilogue on the N-GPU decode path.
            from ..ops import flash

            ext = flash._load_extension()
            if ext is not None:
                out_g, lse_g = ext.combine_packed(gathered, ws, b, h, tq, d)
                return out_g, lse_g
        stacked = gathered.view((ws,) + packed.shape)
        outs = stacked[..., :d]
        lses = stacked[..., d]
        return combine_partials(outs, lses)

    if async_op:
        class _Handle:
            def wait(self):
                work.wait()
                return _finish()

        return _Handle()
    return _finish()


def tree_combine(
    out: torch.Tensor,
    lse: torch.Tensor,
    strategy: str = "auto",
    group: dist.ProcessGroup | None = None,
    async_op: bool = False,
):
    """Dispatch the cross-rank combine.

    strategy:
        "allgather" — one collective, payload * world_size received;
        "allreduce" — two collectives (MAX + packed SUM), bandwidth-optimal;
        "auto"      — allgather when the packed payload is latency-bound
                      (< 1 MiB, i.e. decode), allreduce otherwise (prefill).
    """
    if not (dist.is_available() and dist.is_initialized()) or dist.get_world_size(group) == 1:
        if async_op:
            class _Handle:
                def wait(self):
                    return out.float(), lse.float()

            return _Handle()
        return out.float(), lse.float()
    if strategy == "auto":
        strategy = auto_strategy(out.numel())
    if strategy == "allgather":
        return tree_combine_allgather(out, lse, group=group, async_op=async_op)
    if strategy == "allreduce":
        return tree_combine_allreduce(out, lse, group=group, async_op=async_op)
    raise ValueError(f"unknown combine strategy: {strategy!r}")
