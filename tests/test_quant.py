"""MX fp8 quantization (CPU): group semantics and the outlier-range win
over plain e4m3."""

import torch

from tree_attention_torch_amd.quant import (dequantize_k_mx, mx_group_index,
                                            quantize_k_mx)


def test_group_index_interleaved():
    # silicon-pinned blocks: within each 64-dim window, {0-15, 32-47} and
    # {16-31, 48-63} (tools/check_mx_scales.py)
    g = mx_group_index(torch.arange(128))
    assert g[:16].eq(0).all() and g[32:48].eq(0).all()
    assert g[16:32].eq(1).all() and g[48:64].eq(1).all()
    assert g[64:80].eq(2).all() and g[96:112].eq(2).all()
    assert g[80:96].eq(3).all() and g[112:128].eq(3).all()


def test_unit_variance_is_near_noop():
    torch.manual_seed(0)
    k = torch.randn(2, 4, 64, 128)
    k8, s = quantize_k_mx(k)
    # randn never exceeds 448: every scale is <= 127 and the dequant
    # matches plain fp8 precision class
    assert int(s.max()) <= 127
    dq = dequantize_k_mx(k8, s)
    rel = ((dq - k).abs() / k.abs().clamp(min=1e-6)).median()
    assert rel < 0.04, rel  # e4m3 relative step class


def test_outlier_channels_recovered():
    """The point of MX: outlier channels break plain e4m3 (torch's
    conversion produces NaN above 448 — e4m3fn has no inf); block scales
    recenter the group so the outliers quantize at e4m3 relative
    precision. Small same-group elements lose precision proportional to
    the block max (the inherent MX trade: per-block dynamic range)."""
    torch.manual_seed(1)
    k = torch.randn(1, 2, 256, 128)
    k[..., 7] *= 3000.0   # outlier channel, |values| up to ~10k
    k[..., 40] *= 800.0
    plain = k.to(torch.float8_e4m3fn).float()
    assert torch.isnan(plain).any()  # plain fp8 is BROKEN on this data

    k8, s = quantize_k_mx(k)
    assert torch.isfinite(k8.float()).all()
    dq = dequantize_k_mx(k8, s)
    # outlier channels themselves: full e4m3 relative precision
    for ch in (7, 40):
        rel = ((dq[..., ch] - k[..., ch]).abs() /
               k[..., ch].abs().clamp(min=1e-6)).max()
        assert rel < 0.07, (ch, rel)
    # every element's abs error bounded by the worst e4m3 step of its
    # row's largest block (rel half-step 2^-4 at the bottom of a binade;
    # block max <= row max)
    step = k.abs().amax(dim=-1, keepdim=True) * 2.0 ** -4
    assert ((dq - k).abs() <= step + 1e-6).all()


def test_tiny_values_not_flushed():
    k = torch.full((1, 1, 32, 64), 1e-6)
    k8, s = quantize_k_mx(k)
    dq = dequantize_k_mx(k8, s)
    rel = ((dq - k).abs() / k.abs()).max()
    assert rel < 0.07, rel  # plain e4m3 would flush to subnormals/zero


def test_roundtrip_exact_powers():
    # powers of two within one block's e4m3 span quantize losslessly,
    # including blocks scaled far outside plain e4m3's range
    base = [2.0 ** (i % 8 - 3) for i in range(64)]
    k = torch.tensor([[[base]]])
    k8, s = quantize_k_mx(k)
    torch.testing.assert_close(dequantize_k_mx(k8, s), k)
    k2 = k * 2.0 ** 20   # whole tensor far above plain-fp8 range
    k8b, sb = quantize_k_mx(k2)
    torch.testing.assert_close(dequantize_k_mx(k8b, sb), k2)


def test_quantization_error_bound_property():
    """Hypothesis property: for any finite cache, the MX round-trip error
    is bounded by e4m3's relative precision (2^-3 of the block max after
    E8M0 power-of-two scaling can at most double the spacing) — every
    element within 1/8 of its BLOCK's absmax, zeros preserved."""
    from hypothesis import given, settings, strategies as st

    from tree_attention_torch_amd.quant import dequantize_k_mx, quantize_k_mx

    @settings(max_examples=25, deadline=None)
    @given(st.integers(0, 2**31 - 1), st.floats(-30.0, 30.0))
    def run(seed, log2scale):
        torch.manual_seed(seed)
        k = torch.randn(1, 1, 64, 128, dtype=torch.float32)
        k *= 2.0 ** log2scale
        k8, ks = quantize_k_mx(k)
        kd = dequantize_k_mx(k8, ks)
        assert torch.isfinite(kd).all()
        # per-block bound: group = (d>>6)*2 + ((d>>4)&1) over dims
        d = torch.arange(128)
        g = ((d >> 6) * 2 + ((d >> 4) & 1))
        for blk in range(4):
            cols = g == blk
            sub, subd = k[..., cols], kd[..., cols]
            bound = sub.abs().amax(dim=-1, keepdim=True) / 8.0 + 1e-30
            assert ((sub - subd).abs() <= bound + 1e-6 * sub.abs()).all()
        assert (kd[k == 0] == 0).all()

    run()


def test_v_quantizer_transpose_consistency():
    """quantize_v_mx is quantize_k_mx on the transpose: dequantized V must
    equal the transposed dequantized-K of the transposed input."""
    from tree_attention_torch_amd.quant import (dequantize_k_mx,
                                                dequantize_v_mx,
                                                quantize_k_mx, quantize_v_mx)

    torch.manual_seed(4)
    v = torch.randn(2, 3, 128, 128) * 40.0
    v8, vs = quantize_v_mx(v)
    vd = dequantize_v_mx(v8, vs)
    k8t, kst = quantize_k_mx(v.transpose(-1, -2).contiguous())
    kdt = dequantize_k_mx(k8t, kst).transpose(-1, -2)
    torch.testing.assert_close(vd, kdt)
