"""Oracle correctness: ops/reference.py vs independent formulations.

Covers BASELINE.json config 1 (B=1 H=2 seq=128 d=64 eager CPU) and the
reference-defect regressions of SURVEY.md §0.1 (layout, true LSE, -inf
causal masking).
"""

import math

import pytest
import torch

from tree_attention_torch_amd.ops.reference import (
    attention_reference,
    flash_res_lse,
    repeat_kv,
)


def naive_attention(q, k, v, scale, causal=False, q_offset=0, kv_offset=0):
    """Independent double-precision formulation."""
    q, k, v = q.double(), k.double(), v.double()
    s = q @ k.transpose(-2, -1) * scale
    if causal:
        tq, tk = s.shape[-2], s.shape[-1]
        qpos = torch.arange(q_offset, q_offset + tq)
        kpos = torch.arange(kv_offset, kv_offset + tk)
        s = s.masked_fill(kpos[None, :] > qpos[:, None], float("-inf"))
    p = torch.softmax(s, dim=-1)
    p = torch.nan_to_num(p)  # fully-masked rows
    return p @ v, torch.logsumexp(s, dim=-1)


@pytest.mark.parametrize("b,h,tq,tk,d", [(1, 2, 1, 128, 64), (2, 4, 7, 33, 16)])
def test_matches_naive(b, h, tq, tk, d):
    torch.manual_seed(0)
    q = torch.randn(b, h, tq, d)
    k = torch.randn(b, h, tk, d)
    v = torch.randn(b, h, tk, d)
    scale = 1.0 / math.sqrt(d)
    out, lse = flash_res_lse(q, k, v, scale)
    ref_out, ref_lse = naive_attention(q, k, v, scale)
    torch.testing.assert_close(out, ref_out.float(), rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(lse, ref_lse.float(), rtol=1e-5, atol=1e-5)


def test_matches_sdpa():
    """Cross-check against torch's own scaled_dot_product_attention."""
    torch.manual_seed(1)
    q = torch.randn(1, 2, 128, 64)
    k = torch.randn(1, 2, 128, 64)
    v = torch.randn(1, 2, 128, 64)
    out, _ = flash_res_lse(q, k, v, is_causal=True)
    ref = torch.nn.functional.scaled_dot_product_attention(q, k, v, is_causal=True)
    torch.testing.assert_close(out, ref, rtol=1e-4, atol=1e-4)


def test_true_lse_not_post_softmax():
    """Regression for SURVEY.md §0.1.2: lse must be the pre-softmax LSE."""
    torch.manual_seed(2)
    q = torch.randn(1, 1, 1, 8) * 5
    k = torch.randn(1, 1, 16, 8) * 5
    v = torch.randn(1, 1, 16, 8)
    scale = 1.0
    _, lse = flash_res_lse(q, k, v, scale)
    s = (q.double() @ k.double().transpose(-2, -1) * scale).squeeze()
    expected = torch.logsumexp(s, dim=-1)
    assert abs(lse.item() - expected.item()) < 1e-4
    # the buggy post-softmax value would be in (0, 1+log 16]
    assert not (0 < lse.item() <= 1 + math.log(16)) or abs(expected) < 5


def test_causal_is_neg_inf_not_tril():
    """Regression for SURVEY.md §0.1.3: masked scores contribute nothing."""
    torch.manual_seed(3)
    d = 16
    q = torch.randn(1, 1, 4, d)
    k = torch.randn(1, 1, 4, d)
    v = torch.randn(1, 1, 4, d)
    out, _ = flash_res_lse(q, k, v, 1.0, is_causal=True)
    # row 0 attends only to key 0 => output is exactly v[0]
    torch.testing.assert_close(out[0, 0, 0], v[0, 0, 0].float(), rtol=1e-5, atol=1e-5)


def test_causal_offsets():
    """A query chunk masked against a KV shard via global offsets."""
    torch.manual_seed(4)
    d = 8
    tq, tk = 3, 10
    q = torch.randn(1, 1, tq, d)
    k = torch.randn(1, 1, tk, d)
    v = torch.randn(1, 1, tk, d)
    out, lse = flash_res_lse(q, k, v, 1.0, is_causal=True, q_offset=4, kv_offset=2)
    ref_out, ref_lse = naive_attention(q, k, v, 1.0, True, q_offset=4, kv_offset=2)
    torch.testing.assert_close(out, ref_out.float(), rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(lse, ref_lse.float(), rtol=1e-5, atol=1e-5)


def test_fully_masked_rows():
    """Queries before every key in the shard: out = 0, lse = -inf."""
    q = torch.randn(1, 1, 2, 4)
    k = torch.randn(1, 1, 6, 4)
    v = torch.randn(1, 1, 6, 4)
    out, lse = flash_res_lse(q, k, v, 1.0, is_causal=True, q_offset=0, kv_offset=100)
    assert torch.all(out == 0)
    assert torch.all(torch.isinf(lse)) and torch.all(lse < 0)


def test_gqa_matches_repeated():
    torch.manual_seed(5)
    b, hq, hkv, t, d = 1, 8, 2, 32, 16
    q = torch.randn(b, hq, 1, d)
    k = torch.randn(b, hkv, t, d)
    v = torch.randn(b, hkv, t, d)
    out, lse = flash_res_lse(q, k, v)
    out2, lse2 = flash_res_lse(q, repeat_kv(k, hq // hkv), repeat_kv(v, hq // hkv))
    torch.testing.assert_close(out, out2)
    torch.testing.assert_close(lse, lse2)


def test_half_precision_inputs_close_to_fp32():
    torch.manual_seed(6)
    q = torch.randn(1, 2, 1, 64)
    k = torch.randn(1, 2, 256, 64)
    v = torch.randn(1, 2, 256, 64)
    out32 = attention_reference(q, k, v)
    out16 = attention_reference(q.bfloat16(), k.bfloat16(), v.bfloat16())
    torch.testing.assert_close(out16, out32, rtol=2e-2, atol=2e-2)


def test_driver_main_cpu(monkeypatch, tmp_path):
    """Reference-parity driver main() runs end-to-end on CPU (model.py:129)."""
    import os

    os.chdir(tmp_path)
    from tree_attention_torch_amd.config import TreeAttentionConfig
    from tree_attention_torch_amd.main import main

    cfg = TreeAttentionConfig(seq_len=256, num_heads=2, head_dim=64,
                              dtype="fp32", warmup=1, steps=2)
    main(0, 1, cfg)  # should not raise


def test_bench_contract_cpu(capsys):
    """bench.py emits ONE JSON line with the driver-contract fields."""
    import json

    import bench

    bench.main(["--steps", "2", "--warmup", "1"])
    out = capsys.readouterr().out.strip().splitlines()
    assert len(out) == 1, out
    d = json.loads(out[0])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in d, key
    assert d["metric"] == "attention tokens/sec"
    assert d["n_gpus"] == 1 and d["steps"] == 2 and d["warmup"] == 1
    assert d["scaling"] == "weak" and d["data"] == "synthetic"
    assert d["value"] > 0 and d["ms_per_step"] > 0
    cfg = d["config"]
    assert cfg["num_heads"] == 32 and cfg["head_dim"] == 128
    assert cfg["global_batch"] == 1 and cfg["q_len"] == 1


def test_tree_attention_qlen_exceeds_kv_raises():
    from tree_attention_torch_amd.parallel.tree import tree_attention

    q = torch.randn(1, 2, 16, 32)
    k = torch.randn(1, 2, 8, 32)
    v = torch.randn(1, 2, 8, 32)
    with pytest.raises(ValueError, match="exceeds the global KV length"):
        tree_attention(q, k, v, is_causal=True)


def test_gqa_group_16_oracle():
    """Largest GQA group the decode kernel supports (G=16)."""
    from tree_attention_torch_amd.ops.reference import flash_res_lse

    torch.manual_seed(9)
    q = torch.randn(1, 16, 1, 32)
    k = torch.randn(1, 1, 64, 32)
    v = torch.randn(1, 1, 64, 32)
    out, lse = flash_res_lse(q, k, v)
    ref, _ = flash_res_lse(q, k.expand(1, 16, 64, 32).contiguous(),
                           v.expand(1, 16, 64, 32).contiguous())
    torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-5)
