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
from .parallel.combine import combine_partials, tree_combine

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
        self.mx = isinstance(kv_dtype, str) and kv_dtype in ("mx", "fp8_mx")
        blocks_total = (max_tokens + block - 1) // block + 1
        local_cap = ((blocks_total + self.world - 1) // self.world) * block
        if self.mx:
            # MX block-scaled fp8 cache (quant.py layout): e4m3 payload +
            # per-block E8M0 scales, quantized in closed 64-token WINDOWS
            # (the scale-block granularity). Tokens land in a bf16 staging
            # tail first; each full window quantizes into the main cache.
            # attend() merges the quantized-prefix partial (hardware
            # mfma_scale decode kernel, ZERO-COPY strided views of this
            # preallocated cache) with the tail partial (bf16 kernel) via
            # the standard lse algebra — so per-token accuracy of the
            # newest <=63 tokens is full bf16 and the long prefix streams
            # at fp8 bandwidth with outlier-robust block scales.
            if head_dim != 128:
                raise ValueError("MX KV cache requires head_dim=128")
            if block % 64:
                raise ValueError("MX KV cache requires block % 64 == 0")
            self.k = torch.empty(batch, kv_heads, local_cap, head_dim,
                                 dtype=torch.float8_e4m3fn,
                                 device=self.device)
            self.v = torch.empty_like(self.k)
            self.ks = torch.empty(batch, kv_heads, local_cap, 4,
                                  dtype=torch.uint8, device=self.device)
            self.vs = torch.empty(batch, kv_heads, local_cap // 32,
                                  head_dim, dtype=torch.uint8,
                                  device=self.device)
            self.k_tail = torch.empty(batch, kv_heads, 64, head_dim,
                                      dtype=torch.bfloat16,
                                      device=self.device)
            self.v_tail = torch.empty_like(self.k_tail)
            self.q_len = 0     # quantized prefix length (multiple of 64)
            self.tail_len = 0  # bf16 staging-tail tokens (< 64)
        else:
            td = _DTYPES[kv_dtype] if isinstance(kv_dtype, str) else kv_dtype
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
            if self.mx:
                self._store_mx(k_new, v_new)
            else:
                self.k[:, :, self.local_len : self.local_len + 1] = \
                    k_new.to(self.k.dtype)
                self.v[:, :, self.local_len : self.local_len + 1] = \
                    v_new.to(self.v.dtype)
            self.local_len += 1
        self.total += 1

    def _store_mx(self, kc: torch.Tensor, vc: torch.Tensor) -> None:
        """Store owned tokens (B, Hkv, n, D) into the MX cache: fill the
        bf16 tail, quantize each full 64-token window (bulk windows
        quantize directly when the tail is empty)."""
        from .quant import quantize_k_mx, quantize_v_mx

        n, i = kc.shape[2], 0
        while i < n:
            if self.tail_len > 0 or n - i < 64:
                m = min(64 - self.tail_len, n - i)
                self.k_tail[:, :, self.tail_len : self.tail_len + m] = \
                    kc[:, :, i : i + m].to(torch.bfloat16)
                self.v_tail[:, :, self.tail_len : self.tail_len + m] = \
                    vc[:, :, i : i + m].to(torch.bfloat16)
                self.tail_len += m
                i += m
                if self.tail_len == 64:
                    k8w, ksw = quantize_k_mx(self.k_tail.float())
                    v8w, vsw = quantize_v_mx(self.v_tail.float())
                    q0 = self.q_len
                    self.k[:, :, q0 : q0 + 64] = k8w
                    self.ks[:, :, q0 : q0 + 64] = ksw
                    self.v[:, :, q0 : q0 + 64] = v8w
                    self.vs[:, :, q0 // 32 : q0 // 32 + 2] = vsw
                    self.q_len += 64
                    self.tail_len = 0
            else:
                m = (n - i) // 64 * 64
                # round through bf16 first so bulk-prefilled windows
                # quantize the SAME stream as tail-flushed ones (appends
                # stage in the bf16 tail before quantization)
                k8w, ksw = quantize_k_mx(
                    kc[:, :, i : i + m].to(torch.bfloat16).float())
                v8w, vsw = quantize_v_mx(
                    vc[:, :, i : i + m].to(torch.bfloat16).float())
                q0 = self.q_len
                self.k[:, :, q0 : q0 + m] = k8w
                self.ks[:, :, q0 : q0 + m] = ksw
                self.v[:, :, q0 : q0 + m] = v8w
                self.vs[:, :, q0 // 32 : (q0 + m) // 32] = vsw
                self.q_len += m
                i += m

    def _local_partial_mx(self, q: torch.Tensor,
                          softmax_scale: float | None):
        """MX-cache local partial: quantized-prefix partial (hardware MX
        decode kernel on GPU / dequantized oracle on CPU) merged with the
        bf16 staging-tail partial through the stable lse algebra."""
        parts = []
        if self.q_len > 0:
            if q.device.type == "cuda":
                from .ops.flash import local_attention_mx

                parts.append(local_attention_mx(
                    q, self.k[:, :, : self.q_len],
                    self.ks[:, :, : self.q_len],
                    self.v[:, :, : self.q_len],
                    self.vs[:, :, : self.q_len // 32], softmax_scale))
            else:
                from .quant import dequantize_k_mx, dequantize_v_mx

                kd = dequantize_k_mx(self.k[:, :, : self.q_len].contiguous(),
                                     self.ks[:, :, : self.q_len].contiguous())
                vd = dequantize_v_mx(
                    self.v[:, :, : self.q_len].contiguous(),
                    self.vs[:, :, : self.q_len // 32].contiguous())
                parts.append(local_attention(q.float(), kd, vd,
                                             softmax_scale))
        if self.tail_len > 0:
            kt = self.k_tail[:, :, : self.tail_len].contiguous()
            vt = self.v_tail[:, :, : self.tail_len].contiguous()
            if q.device.type == "cuda":
                parts.append(local_attention(q, kt.to(q.dtype),
                                             vt.to(q.dtype), softmax_scale))
            else:
                parts.append(local_attention(q.float(), kt.float(),
                                             vt.float(), softmax_scale))
        if not parts:
            b, hq, tq, d = q.shape
            out_l = torch.zeros(b, hq, tq, d, dtype=torch.float32,
                                device=q.device)
            lse_l = torch.full((b, hq, tq), float("-inf"),
                               dtype=torch.float32, device=q.device)
            return out_l, lse_l
        if len(parts) == 1:
            return parts[0]
        return combine_partials(torch.stack([p[0] for p in parts]),
                                torch.stack([p[1] for p in parts]))

    def _local_partial(self, q: torch.Tensor, softmax_scale: float | None):
        if self.mx:
            return self._local_partial_mx(q, softmax_scale)
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
        if self.mx:
            raise NotImplementedError(
                "graphed_attend: the MX cache path is two kernels + a "
                "python merge with per-call allocations; use eager "
                "attend() (the MX decode kernel itself is ~0.2 ms at "
                "128K — not launch-bound)")
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
                if self.mx:
                    self._store_mx(k_seq[:, :, t : t + n],
                                   v_seq[:, :, t : t + n])
                else:
                    self.k[:, :, self.local_len : self.local_len + n] = \
                        k_seq[:, :, t : t + n].to(self.k.dtype)
                    self.v[:, :, self.local_len : self.local_len + n] = \
                        v_seq[:, :, t : t + n].to(self.v.dtype)
                self.local_len += n
            self.total += n
            t += n
