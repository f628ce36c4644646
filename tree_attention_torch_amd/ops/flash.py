"""Local attention partial — dispatch between the CDNA4 HIP kernels and the
CPU oracle.

On a GPU (ROCm) device the hand-written gfx950 kernels in ops/hip/ are the
ONLY path: if the compiled extension is missing we raise instead of silently
falling back to eager PyTorch — a silent fallback would fake GPU test passes
without exercising native code. On CPU tensors the fp32 oracle
(ops/reference.py) runs, which is what the no-GPU CI tier exercises.

Reference parity: this layer replaces ``flash_res_lse``'s "simulated" flash
attention (/root/reference/model.py:60-83) with a real flash kernel — the
thing the reference's README admits it never integrated (README.md:21).
"""

from __future__ import annotations

import math
import os

import torch

from . import reference

_EXT = None
_EXT_ERR: Exception | None = None


def _load_extension():
    """Import the in-tree compiled HIP extension (ops/hip/_tree_attn_hip.so)."""
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        from .hip import _tree_attn_hip  # type: ignore

        _EXT = _tree_attn_hip
    except ImportError as e:  # pragma: no cover - GPU-box path
        _EXT_ERR = e
    return _EXT


def hip_available() -> bool:
    return _load_extension() is not None


def decode_loop_chunks(tq: int, g: int, b: int, hq: int) -> int:
    """Spec-decode routing decision (pure; pinned by tests/test_routing.py).

    Returns the number of decode-kernel query chunks to loop over, or 0 when
    a single extension dispatch is right (plain decode Tq*G <= 16 — the
    extension routes that to the split-KV decode kernel — or a prefill-sized
    batch that fills the 512-block grid).

    Cost model (measured sweep, H=32 @ 32K/128K KV): each decode chunk is
    one full-bandwidth KV stream; the underfilled prefill launch costs
    about (512/blocks) streams. Loop while n_chunks is under that.
    Tq=17..32: 0.72 vs 9.62 ms at 128K; Tq=128: 2.65 vs 9.62; crossover
    measured at Tq≈256 (prefill 1.47 vs loop 1.61 at 32K) = the n_chunks
    == 512/blocks point, so strict <.
    """
    if g > 16:
        # MFMA-M batching packs G query heads per KV head into the 16-row
        # M dimension; G > 16 (e.g. MQA Hq=32, Hkv=1) does not fit one tile.
        raise ValueError(
            f"tree_attention: GQA group size Hq/Hkv = {g} exceeds the native "
            "kernel limit of 16 (MFMA-M batching). Replicate KV heads so "
            "that Hq/Hkv <= 16."
        )
    tq_per = 16 // g
    n_chunks = -(-tq // tq_per)
    prefill_blocks = b * hq * (-(-tq // 256))
    max_chunks = max(4, 512 // max(prefill_blocks, 1))
    if tq_per < tq and n_chunks < max_chunks and prefill_blocks < 512:
        return n_chunks
    return 0


def _dispatch_hip(ext, q, k, v, softmax_scale, is_causal, q_offset, kv_offset):
    """Route between the split-KV decode kernel and the prefill kernel.

    The extension itself routes Tq*G <= 16 to decode (MFMA-M batching) and
    everything else to prefill (256-row q-blocks). For SMALL query batches
    just past 16 rows — speculative decode: Tq 17..64 against a long KV —
    the prefill grid is only B*Hq blocks (vs the 512 the chip wants) with
    no KV split, so instead loop the decode kernel over query chunks: each
    chunk launches the full split-KV grid (decode_loop_chunks above).
    """
    tq = q.shape[2]
    g = q.shape[1] // k.shape[1]
    if decode_loop_chunks(tq, g, q.shape[0], q.shape[1]) > 0:
        tq_per = 16 // g
        outs, lses = [], []
        for lo in range(0, tq, tq_per):
            hi = min(tq, lo + tq_per)
            o, l = ext.flash_attention(
                q[..., lo:hi, :].contiguous(), k, v,
                float(softmax_scale), bool(is_causal),
                int(q_offset + lo), int(kv_offset),
            )
            outs.append(o)
            lses.append(l)
        return torch.cat(outs, dim=-2), torch.cat(lses, dim=-1)
    return ext.flash_attention(
        q, k, v, float(softmax_scale), bool(is_causal),
        int(q_offset), int(kv_offset),
    )


def local_attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    softmax_scale: float | None = None,
    is_causal: bool = False,
    q_offset: int = 0,
    kv_offset: int = 0,
) -> tuple[torch.Tensor, torch.Tensor]:
    """Compute the local flash-attention partial (out, lse) on q's device.

    Shapes: q (B, Hq, Tq, D); k, v (B, Hkv, Tk, D) with Hkv | Hq (GQA).
    Returns out (B, Hq, Tq, D) fp32 and lse (B, Hq, Tq) fp32.
    """
    if softmax_scale is None:
        softmax_scale = 1.0 / math.sqrt(q.shape[-1])
    if q.device.type == "cuda":
        ext = _load_extension()
        if ext is None:
            if os.environ.get("TREE_ATTN_ALLOW_EAGER_GPU") == "1":
                return reference.flash_res_lse(
                    q, k, v, softmax_scale, is_causal, q_offset, kv_offset
                )
            raise RuntimeError(
                "tree_attention: the gfx950 HIP extension _tree_attn_hip is not "
                "built but a GPU tensor was passed. Build it with "
                "`python setup.py build_ext --inplace` (or __graft_entry__.build()). "
                f"Import error: {_EXT_ERR}"
            )
        d = q.shape[-1]
        halfish = k.dtype in (torch.bfloat16, torch.float16)
        # D=64 is native for bf16/fp16 (decode and prefill kernels);
        # D in {32,48,80,96,112} is native for bf16/fp16 DECODE (the
        # zero-padded-LDS narrow-head kernel — no global pad copies).
        native_d64 = d == 64 and halfish
        g = q.shape[1] // k.shape[1]
        tq = q.shape[2]
        # will the extension run the prefill kernel (vs decode/spec-loop)?
        will_prefill = tq * g > 16 and \
            decode_loop_chunks(tq, g, q.shape[0], q.shape[1]) == 0
        narrow_decode = (halfish and not will_prefill
                         and d in (32, 48, 80, 96, 112))
        if d < 128 and not native_d64 and not narrow_decode:
            # exact zero-padding fallback: padded dims add 0 to every q.k
            # score and the padded output columns are sliced off.
            # (softmax_scale above was computed from the REAL D.) Pad to the
            # SMALLEST natively supported dim, not always 128.
            import torch.nn.functional as F

            if halfish:
                supported = ((64, 128) if will_prefill
                             else (32, 48, 64, 80, 96, 112, 128))
            else:
                supported = (128,)
            target = next(s for s in supported if s >= d)
            pad = target - d

            def _pad(t):
                if t.dtype == torch.float8_e4m3fn:  # F.pad lacks fp8 support
                    return F.pad(t.view(torch.uint8), (0, pad)).view(
                        torch.float8_e4m3fn).contiguous()
                return F.pad(t, (0, pad)).contiguous()

            out, lse = _dispatch_hip(
                ext, _pad(q), _pad(k), _pad(v),
                softmax_scale, is_causal, q_offset, kv_offset,
            )
            return out[..., :d].contiguous(), lse
        return _dispatch_hip(
            ext, q.contiguous(), k.contiguous(), v.contiguous(),
            softmax_scale, is_causal, q_offset, kv_offset,
        )
    return reference.flash_res_lse(q, k, v, softmax_scale, is_causal, q_offset, kv_offset)


def mx_decode_shaped(hq: int, hkv: int, tq: int) -> bool:
    """MX route decision (pure; pinned by tests/test_routing.py): group
    rows G*Tq <= 16 fit one MFMA-M batch, so the split-KV hardware-scale
    DECODE kernel runs (zero-copy-capable, per-head byte strides);
    anything larger goes to the MX prefill kernel (packed tensors)."""
    return (hq // hkv) * tq <= 16


def _rows_contig(t: "torch.Tensor") -> bool:
    """Per-head rows packed (stride(3)==1, stride(2)==row length) with a
    free head stride — the layout the MX decode kernel binds ZERO-COPY
    (sliced views of a preallocated DecodeSession cache)."""
    return (t.stride(3) == 1 and t.stride(2) == t.shape[3]
            and t.stride(0) == t.stride(1) * t.shape[1])


def local_attention_mx(
    q: torch.Tensor,
    k8: torch.Tensor,
    ks: torch.Tensor,
    v8: torch.Tensor,
    vs: torch.Tensor,
    softmax_scale: float | None = None,
    is_causal: bool = False,
    q_offset: int = 0,
    kv_offset: int = 0,
) -> tuple[torch.Tensor, torch.Tensor]:
    """MX block-scaled fp8 attention: K/V are e4m3 plus per-block E8M0
    scales from quant.quantize_k_mx / quantize_v_mx. The hardware applies
    the scales inside mfma_scale_f32_32x32x64_f8f6f4 (block semantics
    pinned on silicon — tools/check_mx_scales.py), so a KV cache with
    outlier channels/tokens quantizes at full e4m3 relative precision
    where plain fp8 saturates to NaN. GPU-only (no CPU oracle dispatch:
    use quant.dequantize_*_mx + ops.reference for oracles)."""
    if softmax_scale is None:
        softmax_scale = 1.0 / math.sqrt(q.shape[-1])
    ext = _load_extension()
    if ext is None:
        raise RuntimeError("MX fp8 attention requires the HIP extension")
    if q.dtype == torch.float16:
        # the MX kernels quantize Q to e4m3 in-prologue; a bf16 round-trip
        # of an fp16 query is lossless relative to that quantization
        q = q.to(torch.bfloat16)
    decode_shaped = mx_decode_shaped(q.shape[1], k8.shape[1], q.shape[2])
    if not (decode_shaped and all(map(_rows_contig, (k8, ks, v8, vs)))):
        # the prefill kernel wants packed tensors; the decode kernel takes
        # per-head byte strides, so session-cache VIEWS pass zero-copy
        k8, ks = k8.contiguous(), ks.contiguous()
        v8, vs = v8.contiguous(), vs.contiguous()
    return ext.flash_attention_fp8_mx(
        q.contiguous(), k8, ks, v8, vs, float(softmax_scale),
        bool(is_causal), int(q_offset), int(kv_offset),
    )


# Reference-compatible alias: same name/signature shape as model.py:60.
def flash_res_lse(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    softmax_scale: float | None = None,
    is_causal: bool = False,
) -> tuple[torch.Tensor, torch.Tensor]:
    return local_attention(q, k, v, softmax_scale=softmax_scale, is_causal=is_causal)


__all__ = ["local_attention", "local_attention_mx", "flash_res_lse",
           "hip_available", "decode_loop_chunks"]
