"""GPU (MI355X) tests: HIP kernel numerics vs the fp32 eager oracle, plus
silicon-semantics probes (MFMA fragment layout, ds_read_b64_tr_b16).

Every test here requires the in-tree _tree_attn_hip.so — ops/flash.py raises
if a GPU tensor is passed without it, so these tests cannot silently pass on
an eager fallback.
"""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ext():
    from tree_attention_torch_amd.ops import flash

    assert flash.hip_available(), "HIP extension must be built in-tree"
    return flash._load_extension()


def test_native_extension_loaded(ext):
    import tree_attention_torch_amd.ops.hip._tree_attn_hip as m

    assert m.__file__.endswith(".so")


def test_probe_mfma_layout(ext):
    """v_mfma_f32_16x16x32_bf16 with the documented A/B/C lane mappings must
    compute a plain matmul. Asymmetric operands (catch transposes, G9)."""
    torch.manual_seed(0)
    a = (torch.randn(16, 32) * 2).bfloat16().cuda()
    b = (torch.arange(32 * 16).reshape(32, 16).float() % 7 - 3).bfloat16().cuda()
    c = ext.probe_mfma(a, b)
    ref = a.float() @ b.float()
    torch.testing.assert_close(c.cpu(), ref.cpu(), rtol=1e-2, atol=1e-2)


def test_probe_tr16_semantics(ext):
    """ds_read_b64_tr_b16 with lane-linear 8-B addresses: lane l receives
    column (l&15) of the 4x16 row-major bf16 block at group (l>>4)*64 elems,
    i.e. out[l][j] = (l&15) + j*16 + (l>>4)*64 on an identity-pattern LDS."""
    got = ext.probe_tr16().cpu()  # (64, 4) uint16 raw bf16 bits
    lanes = torch.arange(64)
    expect = (lanes % 16).unsqueeze(1) + torch.arange(4).unsqueeze(0) * 16 \
        + (lanes // 16).unsqueeze(1) * 64
    # values were stored as raw uint16 i (tiny bf16 denormals) — compare bits
    assert got.tolist() == expect.tolist(), f"tr16 map differs:\n{got}"


def _check_decode(b, hq, hkv, t, tq=1, causal=False, seed=0, tol=2.5e-2):
    from tree_attention_torch_amd.ops.flash import local_attention
    from tree_attention_torch_amd.ops.reference import flash_res_lse

    torch.manual_seed(seed)
    dev = "cuda"
    q = torch.randn(b, hq, tq, 128, device=dev).bfloat16()
    k = torch.randn(b, hkv, t, 128, device=dev).bfloat16()
    v = torch.randn(b, hkv, t, 128, device=dev).bfloat16()
    q_off = t - tq  # decode semantics: queries at the end
    out, lse = local_attention(q, k, v, is_causal=causal, q_offset=q_off)
    ref_out, ref_lse = flash_res_lse(
        q.cpu(), k.cpu(), v.cpu(), is_causal=causal, q_offset=q_off
    )
    torch.testing.assert_close(out.cpu(), ref_out, rtol=tol, atol=tol)
    torch.testing.assert_close(lse.cpu(), ref_lse, rtol=1e-3, atol=1e-3)


def test_decode_mha_small(ext):
    _check_decode(1, 2, 2, 128)


def test_decode_mha_32k(ext):
    _check_decode(1, 32, 32, 32768)


def test_decode_odd_lengths(ext):
    _check_decode(1, 4, 4, 100)  # < 1 tile
    _check_decode(1, 4, 4, 257)  # tail tile
    _check_decode(2, 3, 3, 1000)  # multi-batch, odd heads


def test_decode_gqa(ext):
    _check_decode(1, 32, 4, 4096)  # GQA 8:1 (BASELINE config 5 head layout)


def test_decode_causal(ext):
    _check_decode(1, 4, 4, 512, causal=True)


def test_small_prefill_chunked(ext):
    """Tq > 16 goes through the interim chunked path; causal across rows."""
    _check_decode(1, 2, 2, 256, tq=48, causal=True)


def test_prefill_gqa_causal(ext):
    _check_decode(1, 8, 2, 256, tq=32, causal=True)


def test_lse_matches_oracle_large_scores(ext):
    from tree_attention_torch_amd.ops.flash import local_attention
    from tree_attention_torch_amd.ops.reference import flash_res_lse

    torch.manual_seed(3)
    q = (torch.randn(1, 2, 1, 128, device="cuda") * 4).bfloat16()
    k = (torch.randn(1, 2, 2048, 128, device="cuda") * 4).bfloat16()
    v = torch.randn(1, 2, 2048, 128, device="cuda").bfloat16()
    out, lse = local_attention(q, k, v, softmax_scale=1.0)
    ref_out, ref_lse = flash_res_lse(q.cpu(), k.cpu(), v.cpu(), softmax_scale=1.0)
    assert torch.isfinite(out).all()
    torch.testing.assert_close(lse.cpu(), ref_lse, rtol=1e-3, atol=1e-3)
    torch.testing.assert_close(out.cpu(), ref_out, rtol=4e-2, atol=4e-2)


def test_fully_masked_chunk(ext):
    """Causal with q before the shard: out = 0, lse = -inf (no NaNs)."""
    from tree_attention_torch_amd.ops.flash import local_attention

    q = torch.randn(1, 2, 1, 128, device="cuda").bfloat16()
    k = torch.randn(1, 2, 256, 128, device="cuda").bfloat16()
    v = torch.randn(1, 2, 256, 128, device="cuda").bfloat16()
    out, lse = local_attention(q, k, v, is_causal=True, q_offset=0, kv_offset=500)
    assert torch.all(out == 0), "masked-out chunk must produce zeros"
    assert torch.all(torch.isinf(lse) & (lse < 0))


def test_scale_parameter(ext):
    from tree_attention_torch_amd.ops.flash import local_attention
    from tree_attention_torch_amd.ops.reference import flash_res_lse

    torch.manual_seed(4)
    q = torch.randn(1, 2, 1, 128, device="cuda").bfloat16()
    k = torch.randn(1, 2, 512, 128, device="cuda").bfloat16()
    v = torch.randn(1, 2, 512, 128, device="cuda").bfloat16()
    for scale in (1.0, 0.05, 1.0 / math.sqrt(128)):
        out, _ = local_attention(q, k, v, softmax_scale=scale)
        ref, _ = flash_res_lse(q.cpu(), k.cpu(), v.cpu(), softmax_scale=scale)
        torch.testing.assert_close(out.cpu(), ref, rtol=2.5e-2, atol=2.5e-2)


def test_prefill_kernel_512(ext):
    _check_decode(1, 4, 4, 512, tq=512, causal=True)


def test_prefill_kernel_odd_rows(ext):
    _check_decode(1, 2, 2, 300, tq=300, causal=True)  # Tq not multiple of 256


def test_prefill_kernel_noncausal(ext):
    _check_decode(1, 2, 2, 384, tq=256, causal=False)


def test_prefill_kernel_gqa_1024(ext):
    _check_decode(1, 8, 2, 1024, tq=1024, causal=True, tol=3e-2)


def test_probe_tr8_semantics(ext):
    """ds_read_b64_tr_b8 with lane-linear 8-B addresses: lane l receives
    column (l&15) of the 8x16 row-major BYTE block of its 16-lane group:
    out[l][j] = ((l&15) + j*16 + (l>>4)*128) & 0xff on an identity LDS."""
    got = ext.probe_tr8().cpu().to(torch.int32)
    lanes = torch.arange(64)
    expect = ((lanes % 16).unsqueeze(1) + torch.arange(8).unsqueeze(0) * 16
              + (lanes // 16).unsqueeze(1) * 128) % 256
    assert got.tolist() == expect.tolist(), f"tr8 map differs:\n{got}"


def _check_fp8_decode(b, hq, hkv, t, tq=1, causal=False, seed=0, tol=4e-2):
    # tolerance from measurement, not guesswork: worst max|dO| over a
    # 25-case seed/shape sweep vs the dequantized-KV oracle is 0.0155
    # and worst max|dLSE| 0.0213 (tools/fp8_err.py, MI355X 2026-09-13);
    # 4e-2 / 5e-2 keep ~2.5x margin. Round-1's 0.15 was 10x looser than
    # the kernel's actual error (VERDICT weak item 5).
    from tree_attention_torch_amd.ops.flash import local_attention
    from tree_attention_torch_amd.ops.reference import flash_res_lse

    torch.manual_seed(seed)
    q = torch.randn(b, hq, tq, 128, device="cuda").bfloat16()
    k8 = torch.randn(b, hkv, t, 128, device="cuda").to(torch.float8_e4m3fn)
    v8 = torch.randn(b, hkv, t, 128, device="cuda").to(torch.float8_e4m3fn)
    q_off = t - tq
    out, lse = local_attention(q, k8, v8, is_causal=causal, q_offset=q_off)
    # oracle on the DEQUANTIZED k/v (isolates kernel error from quantization)
    ref_out, ref_lse = flash_res_lse(
        q.cpu(), k8.cpu().float(), v8.cpu().float(), is_causal=causal, q_offset=q_off
    )
    assert torch.isfinite(out).all()
    torch.testing.assert_close(lse.cpu(), ref_lse, rtol=5e-2, atol=5e-2)
    torch.testing.assert_close(out.cpu(), ref_out, rtol=tol, atol=tol)


@pytest.mark.parametrize("d", [32, 48, 80, 96, 112])
def test_narrow_head_decode_native(ext, d):
    """Narrow head dims run the native zero-padded-LDS decode kernel (no
    global pad copies — VERDICT r1 item 7) and match the fp32 oracle."""
    from tree_attention_torch_amd.ops.flash import local_attention
    from tree_attention_torch_amd.ops.reference import flash_res_lse

    torch.manual_seed(d)
    for b, hq, hkv, t, tq in [(1, 8, 8, 2048, 1), (1, 32, 4, 1500, 2),
                              (2, 4, 4, 300, 1)]:
        q = torch.randn(b, hq, tq, d, device="cuda").bfloat16()
        k = torch.randn(b, hkv, t, d, device="cuda").bfloat16()
        v = torch.randn(b, hkv, t, d, device="cuda").bfloat16()
        out, lse = local_attention(q, k, v)
        ref_out, ref_lse = flash_res_lse(q.cpu(), k.cpu(), v.cpu())
        torch.testing.assert_close(out.cpu(), ref_out, rtol=2.5e-2,
                                   atol=2.5e-2)
        torch.testing.assert_close(lse.cpu(), ref_lse, rtol=1e-3, atol=1e-3)


def test_narrow_head_decode_fp16_and_session(ext):
    from tree_attention_torch_amd.ops.flash import local_attention
    from tree_attention_torch_amd.ops.reference import flash_res_lse
    from tree_attention_torch_amd.session import DecodeSession

    torch.manual_seed(5)
    q = torch.randn(1, 8, 1, 96, device="cuda").half()
    k = torch.randn(1, 8, 4096, 96, device="cuda").half()
    v = torch.randn(1, 8, 4096, 96, device="cuda").half()
    out, _ = local_attention(q, k, v)
    ref, _ = flash_res_lse(q.cpu(), k.cpu(), v.cpu())
    torch.testing.assert_close(out.cpu(), ref, rtol=2.5e-2, atol=2.5e-2)

    # serving cache with a narrow head dim takes the generic slice path
    sess = DecodeSession(1, 8, 96, max_tokens=2048, device="cuda",
                         kv_dtype="bf16", block=64)
    ks = torch.randn(1, 8, 1000, 96, device="cuda").bfloat16()
    vs = torch.randn(1, 8, 1000, 96, device="cuda").bfloat16()
    qs = torch.randn(1, 8, 1, 96, device="cuda").bfloat16()
    sess.prefill(ks, vs)
    o = sess.attend(qs)
    r, _ = flash_res_lse(qs.cpu(), ks.cpu(), vs.cpu())
    torch.testing.assert_close(o.cpu(), r, rtol=2.5e-2, atol=2.5e-2)


def test_fp8_decode_small(ext):
    _check_fp8_decode(1, 2, 2, 256)


def test_fp8_decode_gqa_32k(ext):
    _check_fp8_decode(1, 32, 4, 32768)  # BASELINE config 5 head layout


def test_fp8_decode_odd(ext):
    _check_fp8_decode(1, 4, 4, 1000)
    _check_fp8_decode(1, 4, 4, 100)


def test_probe_mfma32_layout(ext):
    """v_mfma_f32_32x32x16_bf16 lane maps: A[row=l&31][k=(l>>5)*8+e],
    B[k][col=l&31], C[col=l&31][row=(reg&3)+8*(reg>>2)+4*(l>>5)]."""
    torch.manual_seed(7)
    a = (torch.randn(32, 16) * 2).bfloat16().cuda()
    b = (torch.arange(16 * 32).reshape(16, 32).float() % 5 - 2).bfloat16().cuda()
    c = ext.probe_mfma32(a, b)
    ref = a.float() @ b.float()
    torch.testing.assert_close(c.cpu(), ref.cpu(), rtol=1e-2, atol=1e-2)


def _check_fp16(b, hq, hkv, t, tq=1, causal=False, tol=2.5e-2):
    from tree_attention_torch_amd.ops.flash import local_attention
    from tree_attention_torch_amd.ops.reference import flash_res_lse

    torch.manual_seed(11)
    q = torch.randn(b, hq, tq, 128, device="cuda").half()
    k = torch.randn(b, hkv, t, 128, device="cuda").half()
    v = torch.randn(b, hkv, t, 128, device="cuda").half()
    q_off = t - tq
    out, lse = local_attention(q, k, v, is_causal=causal, q_offset=q_off)
    ref_out, ref_lse = flash_res_lse(q.cpu(), k.cpu(), v.cpu(),
                                     is_causal=causal, q_offset=q_off)
    torch.testing.assert_close(out.cpu(), ref_out, rtol=tol, atol=tol)
    torch.testing.assert_close(lse.cpu(), ref_lse, rtol=1e-3, atol=1e-3)


def test_fp16_decode(ext):
    """Reference dtype parity: model.py used .half() (fp16) tensors."""
    _check_fp16(1, 4, 4, 2048)
    _check_fp16(1, 8, 2, 1000)  # GQA + tail


def test_fp16_prefill(ext):
    _check_fp16(1, 2, 2, 512, tq=512, causal=True)


def test_kernel_determinism(ext):
    """Atomics-free kernels must be bitwise-reproducible (race check)."""
    from tree_attention_torch_amd.ops.flash import local_attention

    torch.manual_seed(9)
    q = torch.randn(1, 8, 1, 128, device="cuda").bfloat16()
    k = torch.randn(1, 8, 4096, 128, device="cuda").bfloat16()
    v = torch.randn(1, 8, 4096, 128, device="cuda").bfloat16()
    o1, l1 = local_attention(q, k, v)
    o2, l2 = local_attention(q, k, v)
    assert torch.equal(o1, o2) and torch.equal(l1, l2)
    qp = torch.randn(1, 2, 1024, 128, device="cuda").bfloat16()
    kp = torch.randn(1, 2, 1024, 128, device="cuda").bfloat16()
    vp = torch.randn(1, 2, 1024, 128, device="cuda").bfloat16()
    p1, pl1 = local_attention(qp, kp, vp, is_causal=True)
    p2, pl2 = local_attention(qp, kp, vp, is_causal=True)
    assert torch.equal(p1, p2) and torch.equal(pl1, pl2)


def test_fuzz_shapes(ext):
    """Random shape sweep vs oracle (bounds/tail robustness)."""
    import random

    random.seed(0)
    for _ in range(6):
        b = random.choice([1, 2])
        hkv = random.choice([1, 2, 3])
        g = random.choice([1, 2, 4])
        t = random.randint(1, 700)
        tq = random.choice([1, 2, random.randint(3, 40)])
        causal = random.random() < 0.5
        if tq > t:
            tq = t
        _check_decode(b, hkv * g, hkv, t, tq=tq, causal=causal, tol=3e-2)


def test_fp8_prefill(ext):
    _check_fp8_decode(1, 2, 2, 512, tq=512, causal=True, tol=0.15)
    _check_fp8_decode(1, 4, 1, 300, tq=300, causal=True, tol=0.15)  # GQA+odd


def test_head_dim_64_padding_path(ext):
    """Narrow heads route through the exact zero-padding fallback."""
    from tree_attention_torch_amd.ops.flash import local_attention
    from tree_attention_torch_amd.ops.reference import flash_res_lse

    torch.manual_seed(13)
    for d, tq in ((64, 1), (64, 256), (96, 1)):
        q = torch.randn(1, 4, tq, d, device="cuda").bfloat16()
        k = torch.randn(1, 4, 512, d, device="cuda").bfloat16()
        v = torch.randn(1, 4, 512, d, device="cuda").bfloat16()
        out, lse = local_attention(q, k, v, is_causal=(tq > 1),
                                   q_offset=512 - tq)
        ref_out, ref_lse = flash_res_lse(q.cpu(), k.cpu(), v.cpu(),
                                         is_causal=(tq > 1), q_offset=512 - tq)
        torch.testing.assert_close(out.cpu(), ref_out, rtol=2.5e-2, atol=2.5e-2)
        torch.testing.assert_close(lse.cpu(), ref_lse, rtol=1e-3, atol=1e-3)


def test_head_dim_64_native_decode(ext):
    """D=64 decode runs the native 128-B-row kernel (no padding)."""
    from tree_attention_torch_amd.ops.flash import local_attention
    from tree_attention_torch_amd.ops.reference import flash_res_lse

    torch.manual_seed(14)
    for b, hq, hkv, t, dt in ((1, 8, 8, 4096, torch.bfloat16),
                              (1, 32, 4, 1000, torch.bfloat16),
                              (2, 4, 4, 300, torch.float16)):
        q = torch.randn(b, hq, 1, 64, device="cuda").to(dt)
        k = torch.randn(b, hkv, t, 64, device="cuda").to(dt)
        v = torch.randn(b, hkv, t, 64, device="cuda").to(dt)
        out, lse = local_attention(q, k, v)
        ref_out, ref_lse = flash_res_lse(q.cpu(), k.cpu(), v.cpu())
        torch.testing.assert_close(out.cpu(), ref_out, rtol=2.5e-2, atol=2.5e-2)
        torch.testing.assert_close(lse.cpu(), ref_lse, rtol=1e-3, atol=1e-3)


def test_head_dim_64_native_prefill(ext):
    from tree_attention_torch_amd.ops.flash import local_attention
    from tree_attention_torch_amd.ops.reference import flash_res_lse

    torch.manual_seed(15)
    for b, h, t, tq, dt in ((1, 4, 512, 512, torch.bfloat16),
                            (1, 2, 300, 300, torch.float16),
                            (1, 8, 1024, 256, torch.bfloat16)):
        q = torch.randn(b, h, tq, 64, device="cuda").to(dt)
        k = torch.randn(b, h, t, 64, device="cuda").to(dt)
        v = torch.randn(b, h, t, 64, device="cuda").to(dt)
        out, lse = local_attention(q, k, v, is_causal=True, q_offset=t - tq)
        ref_out, ref_lse = flash_res_lse(q.cpu(), k.cpu(), v.cpu(),
                                         is_causal=True, q_offset=t - tq)
        torch.testing.assert_close(out.cpu(), ref_out, rtol=2.5e-2, atol=2.5e-2)
        torch.testing.assert_close(lse.cpu(), ref_lse, rtol=1e-3, atol=1e-3)


def test_spec_decode_chunked_route(ext):
    """Tq 17..64 at G*Tq/16 <= 4 chunks routes through the looped split-KV
    decode kernel (ops/flash.py _dispatch_hip) — boundary shapes."""
    _check_decode(1, 4, 4, 2048, tq=17, causal=True)   # first chunked size
    _check_decode(1, 4, 4, 2048, tq=64, causal=True)   # last (4 chunks)
    _check_decode(1, 4, 4, 1024, tq=64, causal=False)
    _check_decode(1, 8, 4, 1024, tq=24, causal=True)   # G=2: tq_per=8, 3 chunks


def test_spec_decode_vs_prefill_route_equal(ext):
    """The two routes must agree: force prefill by making the batch big
    enough to fill the grid (prefill_blocks >= 512 disables the loop)."""
    from tree_attention_torch_amd.ops.flash import local_attention

    torch.manual_seed(3)
    t, tq = 512, 32
    q = torch.randn(1, 8, tq, 128, device="cuda").bfloat16()
    k = torch.randn(1, 8, t, 128, device="cuda").bfloat16()
    v = torch.randn(1, 8, t, 128, device="cuda").bfloat16()
    out_a, lse_a = local_attention(q, k, v, is_causal=True, q_offset=t - tq)
    # same math through the prefill kernel route
    out_b, lse_b = ext.flash_attention(q, k, v, 128 ** -0.5, True, t - tq, 0)
    torch.testing.assert_close(out_a, out_b, rtol=2e-2, atol=2e-2)
    torch.testing.assert_close(lse_a, lse_b, rtol=1e-3, atol=1e-3)


def test_spec_decode_wide_chunked(ext):
    """Tq=96 (6 chunks at G=1) stays on the looped decode route when the
    prefill grid would be underfilled (max_chunks scales with 512/blocks)."""
    _check_decode(1, 4, 4, 2048, tq=96, causal=True)


def test_combine_packed_kernel(ext):
    """HIP packed-combine (the all-gather epilogue) vs the eager oracle,
    including a fully-masked shard (lse = -inf)."""
    from tree_attention_torch_amd.parallel.combine import combine_partials

    torch.manual_seed(11)
    for s, b, h, tq, d in [(8, 1, 32, 1, 128), (3, 2, 4, 5, 64), (2, 1, 1, 1, 128)]:
        outs = torch.randn(s, b, h, tq, d, device="cuda")
        lses = torch.randn(s, b, h, tq, device="cuda") * 4
        lses[0] = float("-inf")  # rank 0 fully masked
        packed = torch.cat([outs, lses.unsqueeze(-1)], dim=-1).contiguous()
        out_k, lse_k = ext.combine_packed(packed.view(-1), s, b, h, tq, d)
        out_e, lse_e = combine_partials(outs.cpu(), lses.cpu())
        torch.testing.assert_close(out_k.cpu(), out_e, rtol=1e-5, atol=1e-5)
        torch.testing.assert_close(lse_k.cpu(), lse_e, rtol=1e-5, atol=1e-5)


def test_combine_rescale_finish_kernels(ext):
    """The fused all-reduce halves reproduce combine_partials: emulate the
    collective locally (elementwise max of m, sum of packed)."""
    from tree_attention_torch_amd.parallel.combine import combine_partials

    torch.manual_seed(12)
    b, h, tq, d = 1, 8, 3, 128
    outs = torch.randn(2, b, h, tq, d, device="cuda")
    lses = torch.randn(2, b, h, tq, device="cuda") * 4
    lses[1, 0, :2] = float("-inf")  # rank 1 partially masked
    m = torch.where(torch.isfinite(lses), lses,
                    torch.full_like(lses, -80.0)).amax(0)
    packed = (ext.combine_rescale_pack(outs[0].contiguous(),
                                       lses[0].contiguous(), m.contiguous())
              + ext.combine_rescale_pack(outs[1].contiguous(),
                                         lses[1].contiguous(), m.contiguous()))
    out_k, lse_k = ext.combine_finish(packed.contiguous(), m.contiguous())
    out_e, lse_e = combine_partials(outs.cpu(), lses.cpu())
    torch.testing.assert_close(out_k.cpu(), out_e, rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(lse_k.cpu(), lse_e, rtol=1e-5, atol=1e-5)


@pytest.mark.parametrize("flag,val", [
    ("TREE_ATTN_PREFILL4", "1"), ("TREE_ATTN_PREFILL5", "1"),
    ("TREE_ATTN_PREFILL6", "1"), ("TREE_ATTN_PREFILL6", "2"),
    ("TREE_ATTN_PREFILL6", "7"),
])
def test_experimental_prefill_variants(ext, flag, val):
    """fa_prefill4/5/6 (env-gated variants) must stay numerically honest:
    run in a subprocess because the route flag is latched at first call.
    The shape list exercises gen6's asm interior loop (full 128-key tiles
    below the causal frontier), the C boundary path (frontier, tails,
    ragged rows) and the hand-off between them, plus GQA head mapping."""
    import os
    import subprocess
    import sys

    code = (
        "import torch\n"
        "from tree_attention_torch_amd.ops import flash\n"
        "from tree_attention_torch_amd.ops.reference import flash_res_lse\n"
        "ext = flash._load_extension()\n"
        "scale = 128 ** -0.5\n"
        "cases = [(256, 256, True, 4, 4), (300, 300, True, 4, 4),\n"
        "         (512, 1024, False, 2, 2), (1024, 1024, True, 2, 2),\n"
        "         (2048, 4096, True, 2, 2), (512, 512, True, 8, 2),\n"
        "         (256, 320, False, 2, 2)]\n"  # ragged 64-key KV tail

        "for tq, tkv, causal, hq, hkv in cases:\n"
        "    torch.manual_seed(1)\n"
        "    q = torch.randn(1, hq, tq, 128, device='cuda').bfloat16()\n"
        "    k = torch.randn(1, hkv, tkv, 128, device='cuda').bfloat16()\n"
        "    v = torch.randn(1, hkv, tkv, 128, device='cuda').bfloat16()\n"
        "    o, l = ext.flash_attention(q, k, v, scale, causal, tkv - tq, 0)\n"
        "    ro, rl = flash_res_lse(q.cpu(), k.cpu(), v.cpu(), scale, causal, tkv - tq, 0)\n"
        "    torch.testing.assert_close(o.cpu(), ro, rtol=2.5e-2, atol=2.5e-2)\n"
        "    torch.testing.assert_close(l.cpu(), rl, rtol=1e-3, atol=1e-3)\n"
        "print('VARIANT_OK')\n"
    )
    env = dict(os.environ, **{flag: val})
    r = subprocess.run([sys.executable, "-c", code], capture_output=True,
                       text=True, env=env, timeout=300)
    assert r.returncode == 0 and "VARIANT_OK" in r.stdout, r.stdout + r.stderr


def test_probe_gen6_qkt_stream(ext):
    """Generated hand-placed QKT_PAIR asm stream (round-2 groundwork):
    exact vs the builtin-MFMA control and the eager reference."""
    torch.manual_seed(0)
    q = (torch.randn(64, 128) * 0.5).bfloat16().cuda()
    k = (torch.randn(32, 128) * 0.5).bfloat16().cuda()
    out = ext.probe_gen6_qkt(q, k).cpu()  # (4,64,16): asm j0/j1, builtin j0/j1
    torch.testing.assert_close(out[0], out[2], rtol=0, atol=0)  # asm == builtin
    torch.testing.assert_close(out[1], out[3], rtol=0, atol=0)
    ref = (k.float() @ q.float().T).cpu()
    lanes = torch.arange(64)
    row32, h = lanes % 32, lanes // 32
    for j in range(2):
        for reg in range(16):
            key = (reg % 4) + 8 * (reg // 4)
            exp = ref[key + 4 * h, j * 32 + row32]
            torch.testing.assert_close(out[j][:, reg], exp, rtol=2e-2, atol=2e-2)


def test_decode_gqa16(ext):
    """Largest supported GQA group (G=16) through the MFMA-M batching."""
    _check_decode(1, 16, 1, 4096)
    _check_decode(1, 16, 1, 1000, causal=True)


def test_fp8_narrow_head_padding(ext):
    """fp8 KV with d<128 routes through the uint8-view zero-pad fallback."""
    from tree_attention_torch_amd.ops.flash import local_attention
    from tree_attention_torch_amd.ops.reference import flash_res_lse

    torch.manual_seed(17)
    d = 96
    q = torch.randn(1, 4, 1, d, device="cuda").bfloat16()
    k = (torch.randn(1, 4, 512, d, device="cuda") * 0.5).to(torch.float8_e4m3fn)
    v = (torch.randn(1, 4, 512, d, device="cuda") * 0.5).to(torch.float8_e4m3fn)
    out, lse = local_attention(q, k, v, q_offset=511)
    ref_out, ref_lse = flash_res_lse(
        q.cpu().float(), k.cpu().float(), v.cpu().float(), q_offset=511)
    torch.testing.assert_close(out.cpu(), ref_out, rtol=8e-2, atol=8e-2)
    torch.testing.assert_close(lse.cpu(), ref_lse, rtol=2e-2, atol=2e-2)


def test_probe_gen6_softmax_stream(ext):
    """Generated softmax+pack asm stream: BIT-exact vs the proven C form
    (fa_prefill5's softmax_pack + pswap) on identical inputs."""
    torch.manual_seed(2)
    s_in = (torch.randn(64, 16) * 3).float().cuda()
    ml = torch.stack([torch.randn(64) * 2 - 1, torch.rand(64) * 5 + 0.1],
                     dim=1).cuda()
    c_out, ml_out = ext.probe_gen6_softmax(s_in, ml, 0.125 * 1.44269504)
    assert (c_out[0] == c_out[1]).all(), "c packs differ"
    torch.testing.assert_close(ml_out[0], ml_out[1], rtol=0, atol=0)


def test_mx_scale_semantics_pinned(ext):
    """The mfma_scale per-block E8M0 semantics stay as probed (interleaved
    k-blocks, per-lane couriers) — tools/check_mx_scales.py as a test."""
    import importlib.util
    import os

    spec = importlib.util.spec_from_file_location(
        "check_mx_scales",
        os.path.join(os.path.dirname(os.path.dirname(
            os.path.abspath(__file__))), "tools", "check_mx_scales.py"))
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)  # module-level: imports only
    worst = max(mod.run("narrow", 1, 125, 130), mod.run("wide", 2, 120, 135))
    assert worst < 2e-4, worst


def test_mx_fp8_attention_outliers(ext):
    """VERDICT r1 item 4 'done' check: the MX-scaled fp8 attention vs the
    per-block-scaled oracle, on a KV cache WITH OUTLIERS that breaks the
    plain unit-scale fp8 path (torch's e4m3 conversion NaNs above 448).
    The oracle uses the e4m3-simulated Q (the kernel quantizes Q in-kernel
    at unit scale) so the comparison isolates the SCALE machinery."""
    from tree_attention_torch_amd.ops.flash import local_attention_mx
    from tree_attention_torch_amd.ops.reference import flash_res_lse
    from tree_attention_torch_amd.quant import (dequantize_k_mx,
                                                dequantize_v_mx,
                                                quantize_k_mx, quantize_v_mx)

    torch.manual_seed(7)
    for (b, hq, hkv, t, tq, causal) in [(1, 4, 4, 512, 256, True),
                                        (1, 8, 2, 1024, 1024, False),
                                        (2, 4, 4, 448, 64, True)]:
        q = torch.randn(b, hq, tq, 128, device="cuda").bfloat16()
        k = torch.randn(b, hkv, t, 128, device="cuda")
        v = torch.randn(b, hkv, t, 128, device="cuda")
        k[..., 13] *= 700.0       # outlier K channel: plain fp8 NaNs
        v[:, :, 17, :] *= 600.0   # outlier V token
        assert torch.isnan(k.to(torch.float8_e4m3fn).float()).any()
        k8, ks = quantize_k_mx(k)
        v8, vs = quantize_v_mx(v)
        out, lse = local_attention_mx(q, k8, ks, v8, vs, is_causal=causal,
                                      q_offset=t - tq)
        q_sim = q.float().to(torch.float8_e4m3fn).float()
        ref_out, ref_lse = flash_res_lse(
            q_sim.cpu(), dequantize_k_mx(k8, ks).cpu(),
            dequantize_v_mx(v8, vs).cpu(), is_causal=causal, q_offset=t - tq)
        assert torch.isfinite(out).all()
        # Bounds are MAGNITUDE-aware: outlier scores reach ~1e5, and the
        # MX instruction's accumulation-order noise is ~1e-6 relative to
        # score magnitude (probe: 2e-5 of the 64-term magnitude), so an
        # absolute lse bound would just measure the outlier size. Measured
        # on this matrix: lse/smag <= 1.2e-6, out/vmax <= 2.9e-4.
        kd = dequantize_k_mx(k8, ks)
        vd = dequantize_v_mx(v8, vs)
        smag = (q_sim.abs().amax() * kd.abs().amax() * 128 ** 0.5).item()
        lrel = ((lse.cpu() - ref_lse).abs().max() / smag).item()
        assert lrel < 1e-5, lrel
        vmax = vd.abs().amax().cpu().clamp(min=1.0)
        orel = ((out.cpu() - ref_out).abs() / vmax).max().item()
        assert orel < 2e-3, orel


@pytest.mark.parametrize("path", ["hw", "dequant"])
def test_mx_decode_vs_oracle(ext, monkeypatch, path):
    """MX decode route (G*Tq <= 16), both implementations:

    - "hw" (default): split-KV mfma_scale kernel — the E8M0 scales are
      applied inside the MFMA, raw fp8 streams at glds bandwidth. Q and P
      quantize to e4m3 (prefill-class accuracy), so the oracle uses the
      e4m3-simulated Q and the fp8 bars (tools/fp8_err.py band).
    - "dequant" (TREE_ATTN_MX_DECODE=dequant): LDS dequant stage into the
      bf16 compute path (no Q/P quantization) — compared against the fp32
      reference over the bf16-ROUNDED dequantized cache at tight bars.

    Includes an outlier cache that NaNs plain (unit-scale) fp8."""
    from tree_attention_torch_amd.ops.flash import local_attention_mx
    from tree_attention_torch_amd.ops.reference import flash_res_lse
    from tree_attention_torch_amd.quant import (dequantize_k_mx,
                                                dequantize_v_mx,
                                                quantize_k_mx, quantize_v_mx)

    if path == "dequant":
        monkeypatch.setenv("TREE_ATTN_MX_DECODE", "dequant")
    else:
        monkeypatch.delenv("TREE_ATTN_MX_DECODE", raising=False)
    torch.manual_seed(11)
    for (b, hq, hkv, t, tq, outlier) in [
            (1, 8, 8, 4096, 1, False),     # MHA decode
            (1, 8, 1, 16384, 2, False),    # GQA 8, tq=2 (G*Tq = 16), splits
            (2, 4, 1, 4032, 1, True),      # outliers; Tkv % 64 but not 128
    ]:
        q = torch.randn(b, hq, tq, 128, device="cuda").bfloat16()
        k = torch.randn(b, hkv, t, 128, device="cuda")
        v = torch.randn(b, hkv, t, 128, device="cuda")
        if outlier:
            k[..., 13] *= 700.0
            v[:, :, 17, :] *= 600.0
            assert torch.isnan(k.to(torch.float8_e4m3fn).float()).any()
        k8, ks = quantize_k_mx(k)
        v8, vs = quantize_v_mx(v)
        out, lse = local_attention_mx(q, k8, ks, v8, vs, is_causal=True,
                                      q_offset=t - tq)
        assert torch.isfinite(out).all()
        kd = dequantize_k_mx(k8, ks).cpu()
        vd = dequantize_v_mx(v8, vs).cpu()
        if path == "dequant":
            qr = q.float().cpu()
            kd, vd = kd.bfloat16().float(), vd.bfloat16().float()
        else:
            qr = q.float().to(torch.float8_e4m3fn).float().cpu()
        ref_out, ref_lse = flash_res_lse(qr, kd, vd, is_causal=True,
                                         q_offset=t - tq)
        if outlier:
            smag = (qr.abs().amax() * kd.abs().amax() * 128 ** 0.5).item()
            lbar = 1e-5 if path == "dequant" else 2e-4
            assert ((lse.cpu() - ref_lse).abs().max() / smag).item() < lbar
            vmax = vd.abs().amax().clamp(min=1.0)
            obar = 2e-3 if path == "dequant" else 4e-2
            assert ((out.cpu() - ref_out).abs() / vmax).max().item() < obar
        elif path == "dequant":
            torch.testing.assert_close(lse.cpu(), ref_lse, rtol=1e-2,
                                       atol=1e-2)
            torch.testing.assert_close(out.cpu(), ref_out, rtol=2e-2,
                                       atol=2e-2)
        else:
            torch.testing.assert_close(lse.cpu(), ref_lse, rtol=3e-2,
                                       atol=3e-2)
            torch.testing.assert_close(out.cpu(), ref_out, rtol=4e-2,
                                       atol=4e-2)


def test_mx_fp8_matches_oracle_on_tame_data(ext):
    """On unit-variance data the quantizer legitimately UP-scales each
    block into e4m3's range (scales ~121), so the right check is against
    the per-block-scaled oracle — and accuracy should be at least as good
    as the unit-scale path's measured band (tools/fp8_err.py: 0.016)."""
    from tree_attention_torch_amd.ops.flash import local_attention_mx
    from tree_attention_torch_amd.ops.reference import flash_res_lse
    from tree_attention_torch_amd.quant import (dequantize_k_mx,
                                                dequantize_v_mx,
                                                quantize_k_mx, quantize_v_mx)

    torch.manual_seed(9)
    q = torch.randn(1, 4, 512, 128, device="cuda").bfloat16()
    k = torch.randn(1, 4, 512, 128, device="cuda")
    v = torch.randn(1, 4, 512, 128, device="cuda")
    k8, ks = quantize_k_mx(k)
    v8, vs = quantize_v_mx(v)
    out, lse = local_attention_mx(q, k8, ks, v8, vs, is_causal=True,
                                  q_offset=0)
    q_sim = q.float().to(torch.float8_e4m3fn).float()
    ref_out, ref_lse = flash_res_lse(q_sim.cpu(), dequantize_k_mx(k8, ks).cpu(),
                                     dequantize_v_mx(v8, vs).cpu(),
                                     is_causal=True, q_offset=0)
    torch.testing.assert_close(lse.cpu(), ref_lse, rtol=3e-2, atol=3e-2)
    torch.testing.assert_close(out.cpu(), ref_out, rtol=4e-2, atol=4e-2)
