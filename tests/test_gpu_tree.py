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
