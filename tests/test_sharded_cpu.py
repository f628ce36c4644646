"""Multi-rank-without-multi-GPU: gloo backend, world_size 2/4 on CPU.

The combine collectives are backend-agnostic, so the distributed tree
attention path (parallel/tree.py + parallel/combine.py collective forms) is
exercised here exactly as it runs over RCCL on the GPU node — the mechanism
the reference lacked entirely.
"""

import os

import pytest
import torch
import torch.multiprocessing as mp

from tree_attention_torch_amd.data import make_data
from tree_attention_torch_amd.ops.reference import flash_res_lse


def _worker(rank, world_size, port, strategy, causal, q_len, rep_q):
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    dist.init_process_group("gloo", rank=rank, world_size=world_size)
    try:
        from tree_attention_torch_amd.parallel.tree import tree_attention

        torch.manual_seed(0)
        b, h, d = 1, 4, 32
        t_total = 64 if world_size != 3 else 60
        t_local = t_total // world_size
        # full tensors generated identically on every rank; each rank slices
        # its shard -> ground truth is attention over the full K/V.
        q_full = torch.randn(b, h, q_len, d)
        k_full = torch.randn(b, h, t_total, d)
        v_full = torch.randn(b, h, t_total, d)
        k = k_full[..., rank * t_local : (rank + 1) * t_local, :]
        v = v_full[..., rank * t_local : (rank + 1) * t_local, :]
        out = tree_attention(
            q_full, k, v, is_causal=causal, combine=strategy,
            q_chunk=8 if q_len > 8 else None,
        )
        q_offset = t_total - q_len
        ref, _ = flash_res_lse(q_full, k_full, v_full, is_causal=causal,
                               q_offset=q_offset, kv_offset=0)
        torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-5)
    finally:
        dist.destroy_process_group()


_PORT = [29712]


def _run(world_size, strategy, causal=False, q_len=1, rep_q=True):
    _PORT[0] += 1
    ctx = mp.get_context("spawn")
    procs = [
        ctx.Process(
            target=_worker,
            args=(r, world_size, _PORT[0], strategy, causal, q_len, rep_q),
        )
        for r in range(world_size)
    ]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=120)
    for p in procs:
        assert p.exitcode == 0, f"worker failed with exit code {p.exitcode}"


@pytest.mark.parametrize("strategy", ["allgather", "allreduce", "auto"])
def test_decode_ws2(strategy):
    _run(2, strategy)


def test_decode_ws4():
    _run(4, "auto")


def test_decode_ws3_odd_world():
    # non-power-of-two world: 64 total keys on shards that still divide;
    # exercises the collectives' odd-rank paths
    _run(3, "auto")


@pytest.mark.parametrize("strategy", ["allgather", "allreduce"])
def test_causal_prefill_ws2(strategy):
    # q_len 32 with q_chunk 8 exercises the chunked + overlapped path
    _run(2, strategy, causal=True, q_len=32)


def test_decode_causal_ws2():
    _run(2, "auto", causal=True, q_len=1)


def test_make_data_replicates_q_shards_kv():
    qa, ka, va = make_data((1, 4, 16, 8), rank=0, device="cpu", dtype="fp32")
    qb, kb, vb = make_data((1, 4, 16, 8), rank=1, device="cpu", dtype="fp32")
    torch.testing.assert_close(qa, qb)  # Q replicated
    assert not torch.allclose(ka, kb)  # K sharded via seed
    assert not torch.allclose(va, vb)


def _worker_gqa(rank, world_size, port):
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    dist.init_process_group("gloo", rank=rank, world_size=world_size)
    try:
        from tree_attention_torch_amd.parallel.tree import tree_attention

        torch.manual_seed(7)
        b, hq, hkv, d = 1, 8, 2, 32
        t_total, q_len = 64, 24
        t_local = t_total // world_size
        q = torch.randn(b, hq, q_len, d)
        k_full = torch.randn(b, hkv, t_total, d)
        v_full = torch.randn(b, hkv, t_total, d)
        k = k_full[..., rank * t_local : (rank + 1) * t_local, :]
        v = v_full[..., rank * t_local : (rank + 1) * t_local, :]
        out = tree_attention(q, k, v, is_causal=True, q_chunk=8)
        ref, _ = flash_res_lse(q, k_full, v_full, is_causal=True,
                               q_offset=t_total - q_len)
        torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-5)
    finally:
        dist.destroy_process_group()


def test_gqa_chunked_prefill_ws2():
    _PORT[0] += 1
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker_gqa, args=(r, 2, _PORT[0]))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=120)
    assert all(p.exitcode == 0 for p in procs), [p.exitcode for p in procs]


def _worker_uneven(rank, world_size, port):
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    dist.init_process_group("gloo", rank=rank, world_size=world_size)
    try:
        from tree_attention_torch_amd.parallel.tree import tree_attention

        torch.manual_seed(5)
        b, h, d, t_total, q_len = 1, 4, 32, 96, 8
        bounds = [0, 40, 96]  # rank 0 holds 40 keys, rank 1 holds 56
        q = torch.randn(b, h, q_len, d)
        k_full = torch.randn(b, h, t_total, d)
        v_full = torch.randn(b, h, t_total, d)
        lo, hi = bounds[rank], bounds[rank + 1]
        out = tree_attention(
            q, k_full[..., lo:hi, :], v_full[..., lo:hi, :],
            is_causal=True, kv_offset=lo, total_kv=t_total,
        )
        ref, _ = flash_res_lse(q, k_full, v_full, is_causal=True,
                               q_offset=t_total - q_len)
        torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-5)
    finally:
        dist.destroy_process_group()


def test_uneven_shards_ws2():
    _PORT[0] += 1
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker_uneven, args=(r, 2, _PORT[0]))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=120)
    assert all(p.exitcode == 0 for p in procs), [p.exitcode for p in procs]


def _worker_mx(rank, world_size, port):
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    dist.init_process_group("gloo", rank=rank, world_size=world_size)
    try:
        from tree_attention_torch_amd.parallel.tree import tree_attention
        from tree_attention_torch_amd.quant import (dequantize_k_mx,
                                                    dequantize_v_mx,
                                                    quantize_k_mx,
                                                    quantize_v_mx)

        torch.manual_seed(11)
        b, hkv, d = 1, 2, 128
        t_total = 256
        t_local = t_total // world_size
        q = torch.randn(b, 4, 1, d)
        k_full = torch.randn(b, hkv, t_total, d)
        v_full = torch.randn(b, hkv, t_total, d)
        k_full[..., 9] *= 900.0  # outlier channel: plain fp8 would NaN
        k = k_full[..., rank * t_local:(rank + 1) * t_local, :]
        v = v_full[..., rank * t_local:(rank + 1) * t_local, :]
        k8, ks = quantize_k_mx(k)
        v8, vs = quantize_v_mx(v)
        out = tree_attention(q, k8, v8, kv_scales=(ks, vs))
        # unsharded oracle over the dequantized shards
        kds, vds = [], []
        for r in range(world_size):
            kr = k_full[..., r * t_local:(r + 1) * t_local, :]
            vr = v_full[..., r * t_local:(r + 1) * t_local, :]
            k8r, ksr = quantize_k_mx(kr)
            v8r, vsr = quantize_v_mx(vr)
            kds.append(dequantize_k_mx(k8r, ksr))
            vds.append(dequantize_v_mx(v8r, vsr))
        ref, _ = flash_res_lse(q.float(), torch.cat(kds, dim=-2),
                               torch.cat(vds, dim=-2))
        torch.testing.assert_close(out, ref, rtol=1e-4, atol=1e-4)
    finally:
        dist.destroy_process_group()


def test_mx_sharded_ws2():
    """MX block-scaled fp8 shards through the tree combine (CPU oracle
    path; the GPU kernel path is tests/test_gpu_kernels.py's mx tests)."""
    _PORT[0] += 1
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker_mx, args=(r, 2, _PORT[0]))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=120)
    assert all(p.exitcode == 0 for p in procs), [p.exitcode for p in procs]


def test_mx_kv_offset_causal_cpu():
    """tree_attention(kv_scales=...) with explicit kv_offset/total_kv
    (uneven-shard plumbing) on the CPU oracle path: causal positions are
    GLOBAL, so a middle shard must mask exactly like the dequantized
    oracle with the same offsets."""
    import torch

    from tree_attention_torch_amd.ops.reference import flash_res_lse
    from tree_attention_torch_amd.parallel.tree import tree_attention
    from tree_attention_torch_amd.quant import (dequantize_k_mx,
                                                dequantize_v_mx,
                                                quantize_k_mx, quantize_v_mx)

    torch.manual_seed(21)
    b, h, d = 1, 2, 128
    t_local, kv_off, total = 128, 256, 512
    q = torch.randn(b, h, 64, d).bfloat16()   # last 64 global positions
    k = torch.randn(b, h, t_local, d)
    v = torch.randn(b, h, t_local, d)
    k8, ks = quantize_k_mx(k)
    v8, vs = quantize_v_mx(v)
    out, lse = tree_attention(q, k8, v8, is_causal=True,
                              kv_offset=kv_off, total_kv=total,
                              kv_scales=(ks, vs), return_lse=True)
    ref, ref_lse = flash_res_lse(q.float(), dequantize_k_mx(k8, ks),
                                 dequantize_v_mx(v8, vs), is_causal=True,
                                 q_offset=total - 64, kv_offset=kv_off)
    torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(lse, ref_lse, rtol=1e-5, atol=1e-5)
