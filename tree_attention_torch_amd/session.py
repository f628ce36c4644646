"""Serving-style decode session: a preallocated, sequence-sharded KV cache.

The reference computes attention over freshly-generated random tensors each
run (/root/reference/model.py:145-150); a deployment decodes token by token,
appending each new token's K/V to a cache. This module provides that loop
MI355X-style:

* the cache is preallocated in HBM3E (288 GB/GPU: a 2M-token GQA fp8 cache
  is ~2 GB) and sharded across ranks round-robin by BLOCKS of tokens, so
  every rank's shard stays contiguous for the flash kernel and growth does
  not reshuffle data;
* each decode step appends to exactly ONE rank's shard and runs
  tree_attention over all shards (the stable combine handles ragged shard
  lengths — lse weighting is exact for any split);
* causal correctness across shards comes from per-shard global position
  offsets, maintained here.

Block-cyclic layout: token t lives on rank (t // block) % world at local
position block*((t // block) // world) + t % block.
"""

from __future__ import annotations

import math

import torch
import torch.distributed as dist

from .ops.flash import local_attention
from .parallel.combine import tree_combine

__all__ = ["DecodeSession"]

_DTYPES = {
    "bf16": torch.bfloat16,
    "fp16": torch.float16,
    "fp32": torch.float32,
    "fp8": torch.float8_e4m3fn,
}


class DecodeSession:
    """Token-by-token decode over a sharded, growable KV cache."""

    def __init__(
        self,
        batch: int,
        kv_heads: int,
        head_dim: int,
        max_tokens: int,
        device: torch.device | str = "cuda",
        kv_dtype: str | torch.dtype = "bf16",
        block: int = 256,
        group: dist.ProcessGroup | None = None,
    ) -> None:
        self.device = torch.device(device)
        if dist.is_available() and dist.is_initialized():
            self.rank = dist.get_rank(group)
            self.world = dist.get_world_size(group)
        else:
            self.rank, self.world = 0, 1
        self.group = group
        self.block = block
        td = _DTYPES[kv_dtype] if isinstance(kv_dtype, str) else kv_dtype
        blocks_total = (max_tokens + block - 1) // block + 1
        local_cap = ((blocks_total + self.world - 1) // self.world) * block
        self.k = torch.empty(batch, kv_heads, local_cap, head_dim,
                             dtype=td, device=self.device)
        self.v = torch.empty_like(self.k)
        self.total = 0       # global tokens appended so far
        self.local_len = 0   # tokens in THIS rank's shard

    def _owner(self, t: int) -> int:
        return (t // self.block) % self.world

    def append(self, k_new: torch.Tensor, v_new: torch.Tensor) -> None:
        """Append one token's K/V (B, Hkv, 1, D). Every rank calls this with
        the same tensors; only the owning rank stores them."""
        if self._owner(self.total) == self.rank:
            if self.local_len >= self.k.size(2):
                # without this check the slice assignment below would be an
                # EMPTY slice and a size-1 source legally expands to 0 rows:
                # the append would silently drop the token
                raise RuntimeError(
                    f"DecodeSession capacity exceeded: local shard full at "
                    f"{self.local_len} tokens (grow max_tokens)"
                )
            self.k[:, :, self.local_len : self.local_len + 1] = \
                k_new.to(self.k.dtype)
            self.v[:, :, self.local_len : self.local_len + 1] = \
                v_new.to(self.v.dtype)
            self.local_len += 1
        self.total += 1

    def _local_partial(self, q: torch.Tensor, softmax_scale: float | None):
        # zero-copy eligibility mirrors the binding's TORCH_CHECKs
        # (bindings.cpp flash_attention_cache): D=128; q bf16/fp16; cache
        # dtype == q dtype or fp8 cache under a bf16 q; (Hq/Hkv)*Tq <= 16
        # (MFMA-M batching). Anything else falls through to the generic
        # slice + local_attention path, which handles those cases.
        zero_copy_ok = (
            q.device.type == "cuda"
            and self.k.size(3) == 128
            and q.dtype in (torch.bfloat16, torch.float16)
            and (
                self.k.dtype == q.dtype
                or (self.k.dtype == torch.float8_e4m3fn
                    and q.dtype == torch.bfloat16)
            )
            and (q.shape[1] // self.k.shape[1]) * q.shape[2] <= 16
        )
        if zero_copy_ok:
            # zero-copy cache path: the kernel takes the cache's head stride
            # and reads the LIVE length from a device scalar, so this call
            # is hipGraph-capturable (graphed_attend) and never copies KV.
            from .ops.flash import _load_extension

            ext = _load_extension()
            if ext is not None:
                # NOTE: the length fill lives in sync_len(), NOT here — this
                # function body is hipGraph-captured by graphed_attend, and a
                # captured fill_ would freeze the capture-time length into
                # every replay.
                if not hasattr(self, "_len_dev"):
                    self._len_dev = torch.zeros(1, dtype=torch.long,
                                                device=self.device)
                scale = (softmax_scale if softmax_scale is not None
                         else 1.0 / math.sqrt(q.shape[-1]))
                return ext.flash_attention_cache(q.contiguous(), self.k,
                                                 self.v, self._len_dev, scale)
        if self.local_len > 0:
            k = self.k[:, :, : self.local_len].contiguous()
            v = self.v[:, :, : self.local_len].contiguous()
            return local_attention(q, k, v, softmax_scale)
        b, hq, tq, d = q.shape
        out_l = torch.zeros(b, hq, tq, d, dtype=torch.float32, device=q.device)
        lse_l = torch.full((b, hq, tq), float("-inf"), dtype=torch.float32,
                           device=q.device)
        return out_l, lse_l

    def attend(self, q: torch.Tensor, softmax_scale: float | None = None,
               combine: str = "auto") -> torch.Tensor:
        """Attention of q (B, Hq, 1, D) over every cached token.

        All cached tokens precede the query (decode semantics), so no causal
        mask is needed; the block-cyclic placement therefore needs no
        position bookkeeping inside the kernel — only the shard contents
        matter, and the combine is permutation-invariant.
        """
        assert self.total > 0, "attend() before any append()"
        self.sync_len()
        out_l, lse_l = self._local_partial(q, softmax_scale)
        out, _ = tree_combine(out_l, lse_l, strategy=combine, group=self.group)
        return out

    def graphed_attend(self, q_static: torch.Tensor,
                       softmax_scale: float | None = None,
                       combine: str = "auto"):
        """Capture the decode-attend into a hipGraph.

        Returns (replay, out): write the query into q_static, keep
        self._len_dev fresh via sync_len(), call replay() -> output tensor.
        The kernel reads the live KV length from device memory, so ONE
        captured graph serves the whole growing sequence — no recapture.

        world_size 1: the whole step replays from the graph and ``out`` is
        the static output buffer. world_size > 1: the LOCAL partial replays
        from the graph (launch-bound part of the step) and the cross-rank
        combine collective runs eagerly per call — RCCL collectives are not
        capturable into a per-rank local graph, but they are a single
        latency-bound call on a ~16 KB payload; ``out`` is None and
        replay()'s return value is the fresh combined output.
        """
        assert self.device.type == "cuda"
        # warmup on a side stream (allocator + kernels), then capture
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                self._local_partial(q_static, softmax_scale)
        torch.cuda.current_stream().wait_stream(s)
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph):
            out_l, lse_l = self._local_partial(q_static, softmax_scale)
        self._graph = graph  # keep alive

        if self.world == 1:
            def replay() -> torch.Tensor:
                graph.replay()
                return out_l

            return replay, out_l

        def replay_dist() -> torch.Tensor:
            graph.replay()
            out, _ = tree_combine(out_l, lse_l, strategy=combine,
                                  group=self.group)
            return out

        return replay_dist, None

    def sync_len(self) -> None:
        """Refresh the device-side length after append()s (graphed path)."""
        if hasattr(self, "_len_dev"):
            self._len_dev.fill_(self.local_len)
        elif self.device.type == "cuda":
            self._len_dev = torch.full((1,), self.local_len, dtype=torch.long,
                                       device=self.device)

    def prefill(self, k_seq: torch.Tensor, v_seq: torch.Tensor) -> None:
        """Bulk-append a prompt's K/V (B, Hkv, T, D), preserving the
        block-cyclic ownership (chunks split at global block boundaries)."""
        t_total = k_seq.shape[2]
        t = 0
        while t < t_total:
            n = min(self.block - self.total % self.block, t_total - t)
            if self._owner(self.total) == self.rank:
                if self.local_len + n > self.k.size(2):
                    raise RuntimeError(
                        f"DecodeSession capacity exceeded during prefill at "
                        f"{self.local_len}+{n} tokens (grow max_tokens)"
                    )
                self.k[:, :, self.local_len : self.local_len + n] = \
                    k_seq[:, :, t : t + n].to(self.k.dtype)
                self.v[:, :, self.local_len : self.local_len + n] = \
                    v_seq[:, :, t : t + n].to(self.v.dtype)
                self.local_len += n
            self.total += n
            t += n
