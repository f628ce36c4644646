"""GPU end-to-end: tree_attention on a single rank (world_size 1) must equal
the fp32 oracle; multi-rank GPU equality is covered by the driver's 8-GPU
round-end run (the collective math itself is validated by the gloo tests)."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_tree_attention_single_gpu():
    import tree_attention_torch_amd as ta

    torch.manual_seed(0)
    q, k, v = ta.make_data((1, 32, 8192, 128), rank=0, device="cuda", dtype="bf16")
    out = ta.tree_attention(q, k, v)
    ref = ta.attention_reference(q.cpu(), k.cpu(), v.cpu())
    torch.testing.assert_close(out.cpu(), ref, rtol=2.5e-2, atol=2.5e-2)


def test_tree_decode_reference_surface():
    import tree_attention_torch_amd as ta

    q, k, v = ta.make_data((1, 16, 2048, 128), rank=0, device="cuda", dtype="bf16")
    out = ta.tree_decode(q, k, v, rank=0, world_size=1, device=torch.device("cuda"))
    assert out.shape == (1, 16, 1, 128)
    assert torch.isfinite(out).all()


def test_gqa_tree_attention_gpu():
    import tree_attention_torch_amd as ta

    torch.manual_seed(1)
    q, k, v = ta.make_data(
        (1, 32, 4096, 128), rank=0, device="cuda", dtype="bf16", kv_heads=4
    )
    out = ta.tree_attention(q, k, v)
    ref = ta.attention_reference(q.cpu(), k.cpu(), v.cpu())
    torch.testing.assert_close(out.cpu(), ref, rtol=2.5e-2, atol=2.5e-2)


def test_fp8_gqa_tree_attention_gpu():
    """BASELINE config 5 path end-to-end: fp8 KV + GQA through tree_attention."""
    import tree_attention_torch_amd as ta

    torch.manual_seed(2)
    q, k, v = ta.make_data(
        (1, 32, 8192, 128), rank=0, device="cuda", dtype="fp8", kv_heads=4
    )
    out = ta.tree_attention(q, k, v)
    ref = ta.attention_reference(q.cpu(), k.cpu().float(), v.cpu().float())
    torch.testing.assert_close(out.cpu(), ref, rtol=0.15, atol=0.15)


def test_chunked_prefill_tree_attention_gpu():
    """Chunked causal prefill through tree_attention (single rank): the
    q_chunk path must agree with one-shot prefill and the oracle."""
    import tree_attention_torch_amd as ta

    torch.manual_seed(3)
    tq = 2048
    q, k, v = ta.make_data((1, 8, tq, 128), rank=0, device="cuda",
                           q_len=tq, dtype="bf16")
    out_chunked = ta.tree_attention(q, k, v, is_causal=True, q_chunk=512)
    out_oneshot = ta.tree_attention(q, k, v, is_causal=True, q_chunk=tq)
    torch.testing.assert_close(out_chunked, out_oneshot, rtol=1e-4, atol=1e-4)
    ref = ta.attention_reference(q.cpu(), k.cpu(), v.cpu(), is_causal=True)
    torch.testing.assert_close(out_chunked.cpu(), ref, rtol=3e-2, atol=3e-2)


def test_tree_attention_mx_kv_scales_gpu():
    """tree_attention(kv_scales=...) routes the MX hardware kernel: at
    world 1 the result must BE local_attention_mx's output (trivial
    combine). Numerics-vs-oracle for the MX kernel live in
    test_gpu_kernels.py (magnitude-aware bounds — outlier scores amplify
    accumulation noise beyond naive tolerances)."""
    import torch

    from tree_attention_torch_amd.ops.flash import local_attention_mx
    from tree_attention_torch_amd.parallel.tree import tree_attention
    from tree_attention_torch_amd.quant import quantize_k_mx, quantize_v_mx

    torch.manual_seed(3)
    q = torch.randn(1, 4, 64, 128, device="cuda").bfloat16()
    k = torch.randn(1, 4, 512, 128, device="cuda")
    k[..., 30] *= 800.0
    v = torch.randn(1, 4, 512, 128, device="cuda")
    k8, ks = quantize_k_mx(k)
    v8, vs = quantize_v_mx(v)
    out = tree_attention(q, k8, v8, is_causal=True, kv_scales=(ks, vs))
    ref, _ = local_attention_mx(q, k8, ks, v8, vs, is_causal=True,
                                q_offset=448)
    torch.testing.assert_close(out, ref.float())
