"""Properties of the stable (out, lse) combine (parallel/combine.py).

The combine must be exact (sharded == unsharded), associative/commutative
(order-independent — the property that makes all-reduce a valid tree
reduction AND doubles as the race detector for the collective path,
SURVEY.md §5.2), and robust to -inf LSEs from fully-masked shards.
"""

import torch

from tree_attention_torch_amd.ops.reference import flash_res_lse
from tree_attention_torch_amd.parallel.combine import combine_partials


def _shard_partials(q, k, v, n_shards, scale=None, causal=False, q_offset=0):
    t = k.shape[-2]
    assert t % n_shards == 0
    tl = t // n_shards
    outs, lses = [], []
    for r in range(n_shards):
        kr = k[..., r * tl : (r + 1) * tl, :]
        vr = v[..., r * tl : (r + 1) * tl, :]
        o, l = flash_res_lse(q, kr, vr, scale, causal, q_offset=q_offset, kv_offset=r * tl)
        outs.append(o)
        lses.append(l)
    return torch.stack(outs), torch.stack(lses)


def test_sharded_equals_unsharded():
    torch.manual_seed(0)
    b, h, t, d = 2, 4, 64, 32
    q = torch.randn(b, h, 1, d)
    k = torch.randn(b, h, t, d)
    v = torch.randn(b, h, t, d)
    ref_out, ref_lse = flash_res_lse(q, k, v)
    for s in (2, 4, 8):
        outs, lses = _shard_partials(q, k, v, s)
        out, lse = combine_partials(outs, lses)
        torch.testing.assert_close(out, ref_out, rtol=1e-5, atol=1e-5)
        torch.testing.assert_close(lse, ref_lse, rtol=1e-5, atol=1e-5)


def test_order_independence():
    """Any permutation / association of shards gives the same answer."""
    torch.manual_seed(1)
    q = torch.randn(1, 2, 1, 16)
    k = torch.randn(1, 2, 32, 16)
    v = torch.randn(1, 2, 32, 16)
    outs, lses = _shard_partials(q, k, v, 8)
    base_out, base_lse = combine_partials(outs, lses)
    for perm in (torch.randperm(8), torch.randperm(8), torch.arange(7, -1, -1)):
        out_p, lse_p = combine_partials(outs[perm], lses[perm])
        torch.testing.assert_close(out_p, base_out, rtol=1e-5, atol=1e-6)
        torch.testing.assert_close(lse_p, base_lse, rtol=1e-5, atol=1e-6)
    # associativity: combine pairwise tree vs flat
    o01, l01 = combine_partials(outs[:4], lses[:4])
    o23, l23 = combine_partials(outs[4:], lses[4:])
    out_tree, lse_tree = combine_partials(
        torch.stack([o01, o23]), torch.stack([l01, l23])
    )
    torch.testing.assert_close(out_tree, base_out, rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(lse_tree, base_lse, rtol=1e-5, atol=1e-6)


def test_causal_sharded_prefill():
    """Causal masking across shards: later shards are fully masked for early
    queries (-inf lse) and the combine must still be exact."""
    torch.manual_seed(2)
    b, h, t, d = 1, 2, 32, 8
    q = torch.randn(b, h, t, d)
    k = torch.randn(b, h, t, d)
    v = torch.randn(b, h, t, d)
    ref_out, ref_lse = flash_res_lse(q, k, v, is_causal=True)
    outs, lses = _shard_partials(q, k, v, 4, causal=True, q_offset=0)
    assert torch.isinf(lses).any(), "test must exercise fully-masked shards"
    out, lse = combine_partials(outs, lses)
    torch.testing.assert_close(out, ref_out, rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(lse, ref_lse, rtol=1e-5, atol=1e-5)


def test_extreme_logits_stable():
    """Large-magnitude scores must not overflow the combine."""
    torch.manual_seed(3)
    q = torch.randn(1, 1, 1, 8) * 30
    k = torch.randn(1, 1, 64, 8) * 30
    v = torch.randn(1, 1, 64, 8)
    outs, lses = _shard_partials(q, k, v, 8, scale=1.0)
    out, lse = combine_partials(outs, lses)
    ref_out, ref_lse = flash_res_lse(q, k, v, 1.0)
    assert torch.isfinite(out).all() and torch.isfinite(lse).all()
    torch.testing.assert_close(out, ref_out, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(lse, ref_lse, rtol=1e-4, atol=1e-4)


def test_single_shard_identity():
    torch.manual_seed(4)
    q = torch.randn(1, 2, 3, 16)
    k = torch.randn(1, 2, 20, 16)
    v = torch.randn(1, 2, 20, 16)
    o, l = flash_res_lse(q, k, v)
    out, lse = combine_partials(o.unsqueeze(0), l.unsqueeze(0))
    torch.testing.assert_close(out, o)
    torch.testing.assert_close(lse, l)


def test_combine_partials_property_random_splits():
    """Hypothesis property: combining ANY partition of the KV into S
    partials reproduces the joint attention (the algebra is exact up to
    fp32 roundoff for every split arrangement, not just equal shards)."""
    from hypothesis import given, settings, strategies as st

    from tree_attention_torch_amd.ops.reference import flash_res_lse
    from tree_attention_torch_amd.parallel.combine import combine_partials

    @settings(max_examples=20, deadline=None)
    @given(st.integers(0, 2**31 - 1),
           st.lists(st.integers(1, 64), min_size=1, max_size=6))
    def run(seed, sizes):
        torch.manual_seed(seed)
        t = sum(sizes)
        q = torch.randn(1, 2, 1, 32)
        k = torch.randn(1, 2, t, 32)
        v = torch.randn(1, 2, t, 32)
        ref, ref_lse = flash_res_lse(q, k, v)
        outs, lses = [], []
        lo = 0
        for n in sizes:
            o, l = flash_res_lse(q, k[:, :, lo : lo + n],
                                 v[:, :, lo : lo + n])
            outs.append(o)
            lses.append(l)
            lo += n
        out, lse = combine_partials(torch.stack(outs), torch.stack(lses))
        torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-5)
        torch.testing.assert_close(lse, ref_lse, rtol=1e-5, atol=1e-5)

    run()


def test_combine_partials_all_empty():
    """All partials empty (-inf lse): out = 0, lse = -inf, no NaN."""
    from tree_attention_torch_amd.parallel.combine import combine_partials

    outs = torch.zeros(3, 1, 2, 1, 16)
    lses = torch.full((3, 1, 2, 1), float("-inf"))
    out, lse = combine_partials(outs, lses)
    assert torch.isfinite(out).all() and (out == 0).all()
    assert torch.isinf(lse).all() and (lse < 0).all()
