"""DecodeSession: sharded KV-cache decode loop vs the full-attention oracle."""

import os

import pytest
import torch
import torch.multiprocessing as mp

from tree_attention_torch_amd.ops.reference import flash_res_lse
from tree_attention_torch_amd.session import DecodeSession


def test_session_single_rank_cpu():
    torch.manual_seed(0)
    b, h, d = 1, 4, 32
    sess = DecodeSession(b, h, d, max_tokens=512, device="cpu",
                         kv_dtype="fp32", block=8)
    ks = torch.randn(b, h, 40, d)
    vs = torch.randn(b, h, 40, d)
    sess.prefill(ks[:, :, :17], vs[:, :, :17])  # non-aligned prompt
    for t in range(17, 40):
        sess.append(ks[:, :, t : t + 1], vs[:, :, t : t + 1])
        q = torch.randn(b, h, 1, d)
        out = sess.attend(q)
        ref, _ = flash_res_lse(q, ks[:, :, : t + 1], vs[:, :, : t + 1])
        torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-5)


def _worker(rank, world, port):
    import torch.distributed as dist

    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        torch.manual_seed(0)
        b, h, d = 1, 2, 16
        sess = DecodeSession(b, h, d, max_tokens=256, device="cpu",
                             kv_dtype="fp32", block=4)
        ks = torch.randn(b, h, 30, d)
        vs = torch.randn(b, h, 30, d)
        sess.prefill(ks[:, :, :10], vs[:, :, :10])
        for t in range(10, 30):
            sess.append(ks[:, :, t : t + 1], vs[:, :, t : t + 1])
            q = torch.randn(b, h, 1, d)
            out = sess.attend(q)
            ref, _ = flash_res_lse(q, ks[:, :, : t + 1], vs[:, :, : t + 1])
            torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-5)
    finally:
        dist.destroy_process_group()


def test_session_sharded_ws2():
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker, args=(r, 2, 29741)) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=120)
    assert all(p.exitcode == 0 for p in procs), [p.exitcode for p in procs]


@pytest.mark.gpu
def test_session_gpu_bf16():
    torch.manual_seed(1)
    b, h, d = 1, 8, 128
    sess = DecodeSession(b, h, d, max_tokens=4096, device="cuda",
                         kv_dtype="bf16", block=256)
    ks = torch.randn(b, h, 2000, d, device="cuda").bfloat16()
    vs = torch.randn(b, h, 2000, d, device="cuda").bfloat16()
    sess.prefill(ks[:, :, :1999], vs[:, :, :1999])
    sess.append(ks[:, :, 1999:2000], vs[:, :, 1999:2000])
    q = torch.randn(b, h, 1, d, device="cuda").bfloat16()
    out = sess.attend(q)
    ref, _ = flash_res_lse(q.cpu(), ks.cpu(), vs.cpu())
    torch.testing.assert_close(out.cpu(), ref, rtol=2.5e-2, atol=2.5e-2)


@pytest.mark.gpu
def test_session_graphed_decode():
    """hipGraph-captured decode step keeps working as the cache grows."""
    torch.manual_seed(2)
    b, h, d = 1, 8, 128
    sess = DecodeSession(b, h, d, max_tokens=2048, device="cuda",
                         kv_dtype="bf16", block=256)
    ks = torch.randn(b, h, 1024, d, device="cuda").bfloat16()
    vs = torch.randn(b, h, 1024, d, device="cuda").bfloat16()
    sess.prefill(ks[:, :, :512], vs[:, :, :512])
    q_static = torch.zeros(b, h, 1, d, device="cuda").bfloat16()
    sess.sync_len()
    replay, out_static = sess.graphed_attend(q_static)
    for t in (512, 700, 1023):
        sess.prefill(ks[:, :, sess.total : t], vs[:, :, sess.total : t])
        sess.append(ks[:, :, t : t + 1], vs[:, :, t : t + 1])
        sess.sync_len()
        q = torch.randn(b, h, 1, d, device="cuda").bfloat16()
        q_static.copy_(q)
        out = replay()
        ref, _ = flash_res_lse(q.cpu(), ks[:, :, : t + 1].cpu(),
                               vs[:, :, : t + 1].cpu())
        torch.testing.assert_close(out.cpu(), ref, rtol=2.5e-2, atol=2.5e-2)


@pytest.mark.gpu
def test_session_head_dim_64():
    """Non-128 head dims fall back from the cache binding to the generic
    (native-D64 / padded) path."""
    torch.manual_seed(4)
    b, h, d = 1, 4, 64
    sess = DecodeSession(b, h, d, max_tokens=1024, device="cuda",
                         kv_dtype="bf16", block=128)
    ks = torch.randn(b, h, 600, d, device="cuda").bfloat16()
    vs = torch.randn(b, h, 600, d, device="cuda").bfloat16()
    sess.prefill(ks, vs)
    q = torch.randn(b, h, 1, d, device="cuda").bfloat16()
    out = sess.attend(q)
    ref, _ = flash_res_lse(q.cpu(), ks.cpu(), vs.cpu())
    torch.testing.assert_close(out.cpu(), ref, rtol=2.5e-2, atol=2.5e-2)


def test_session_capacity_overflow_raises():
    sess = DecodeSession(1, 2, 16, max_tokens=8, device="cpu",
                         kv_dtype="fp32", block=4)
    k1 = torch.randn(1, 2, 1, 16)
    for _ in range(12):  # local_cap = 3 blocks of 4 = 12
        sess.append(k1, k1)
    with pytest.raises(RuntimeError, match="capacity exceeded"):
        for _ in range(8):
            sess.append(k1, k1)


def test_session_batched_cpu():
    """Batched serving (B=3): the cache and attend are batch-dim clean."""
    torch.manual_seed(4)
    b, h, d = 3, 4, 32
    sess = DecodeSession(b, h, d, max_tokens=256, device="cpu",
                         kv_dtype="fp32", block=8)
    ks = torch.randn(b, h, 33, d)
    vs = torch.randn(b, h, 33, d)
    sess.prefill(ks[:, :, :9], vs[:, :, :9])
    for t in range(9, 33):
        sess.append(ks[:, :, t : t + 1], vs[:, :, t : t + 1])
        q = torch.randn(b, h, 1, d)
        out = sess.attend(q)
        ref, _ = flash_res_lse(q, ks[:, :, : t + 1], vs[:, :, : t + 1])
        torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-5)


@pytest.mark.gpu
def test_session_batched_gpu():
    """Batched serving on the zero-copy cache kernel path (B=4, bf16)."""
    torch.manual_seed(5)
    b, h, d = 4, 8, 128
    sess = DecodeSession(b, h, d, max_tokens=4096, device="cuda",
                         kv_dtype="bf16", block=256)
    ks = torch.randn(b, h, 600, d, device="cuda").bfloat16()
    vs = torch.randn(b, h, 600, d, device="cuda").bfloat16()
    sess.prefill(ks[:, :, :512], vs[:, :, :512])
    for t in range(512, 600, 17):
        q = torch.randn(b, h, 1, d, device="cuda").bfloat16()
        out = sess.attend(q)
        ref, _ = flash_res_lse(q.cpu(), ks[:, :, :sess.total].cpu(),
                               vs[:, :, :sess.total].cpu())
        torch.testing.assert_close(out.cpu(), ref, rtol=2.5e-2, atol=2.5e-2)
        for u in range(t, min(t + 17, 600)):
            sess.append(ks[:, :, u : u + 1], vs[:, :, u : u + 1])


def test_session_mx_cpu():
    """MX-cache session vs the dequantized-cache oracle on CPU: the
    quantized prefix (closed 64-token windows) + bf16 staging tail must
    together attend exactly like the dequantized union."""
    from tree_attention_torch_amd.quant import (dequantize_k_mx,
                                                dequantize_v_mx)

    torch.manual_seed(3)
    b, h, d = 1, 2, 128
    sess = DecodeSession(b, h, d, max_tokens=512, device="cpu",
                         kv_dtype="mx", block=64)
    t_all = 200
    ks = torch.randn(b, h, t_all, d)
    vs = torch.randn(b, h, t_all, d)
    ks[..., 13] *= 700.0  # outlier channel: plain fp8 would NaN
    sess.prefill(ks[:, :, :150], vs[:, :, :150])
    assert sess.q_len == 128 and sess.tail_len == 22
    for t in range(150, t_all):
        sess.append(ks[:, :, t : t + 1], vs[:, :, t : t + 1])
        q = torch.randn(b, h, 1, d)
        out = sess.attend(q)
        # oracle: dequantized prefix + bf16 tail, matching the storage
        n = t + 1
        qn = n // 64 * 64
        from tree_attention_torch_amd.quant import (quantize_k_mx,
                                                     quantize_v_mx)

        kd = dequantize_k_mx(
            *quantize_k_mx(ks[:, :, :qn].bfloat16().float()))
        vd = dequantize_v_mx(
            *quantize_v_mx(vs[:, :, :qn].bfloat16().float()))
        ku = torch.cat([kd, ks[:, :, qn:n].bfloat16().float()], dim=2)
        vu = torch.cat([vd, vs[:, :, qn:n].bfloat16().float()], dim=2)
        ref, _ = flash_res_lse(q.float(), ku, vu)
        torch.testing.assert_close(out, ref, rtol=1e-4, atol=1e-4)
        assert torch.isfinite(out).all()


def test_session_mx_validation():
    with pytest.raises(ValueError, match="head_dim"):
        DecodeSession(1, 2, 64, max_tokens=128, device="cpu",
                      kv_dtype="mx", block=64)
    with pytest.raises(ValueError, match="block"):
        DecodeSession(1, 2, 128, max_tokens=128, device="cpu",
                      kv_dtype="mx", block=50)


def _worker_mx_session(rank, world, port):
    import torch.distributed as dist

    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        torch.manual_seed(5)
        b, h, d = 1, 2, 128
        sess = DecodeSession(b, h, d, max_tokens=1024, device="cpu",
                             kv_dtype="mx", block=64)
        t_all = 300
        ks = torch.randn(b, h, t_all, d)
        vs = torch.randn(b, h, t_all, d)
        sess.prefill(ks[:, :, :260], vs[:, :, :260])
        for t in range(260, t_all):
            sess.append(ks[:, :, t : t + 1], vs[:, :, t : t + 1])
            q = torch.randn(b, h, 1, d)
            out = sess.attend(q)
            # cross-rank union must stay finite and close to the bf16
            # full-precision oracle within MX quantization error
            ref, _ = flash_res_lse(q.float(), ks[:, :, : t + 1].float(),
                                   vs[:, :, : t + 1].float())
            assert torch.isfinite(out).all()
            torch.testing.assert_close(out, ref, rtol=5e-2, atol=5e-2)
    finally:
        dist.destroy_process_group()


def test_session_mx_sharded_ws2():
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker_mx_session, args=(r, 2, 29747))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=240)
    assert all(p.exitcode == 0 for p in procs), [p.exitcode for p in procs]


@pytest.mark.gpu
def test_session_mx_gpu():
    """MX-cache session on silicon: quantized prefix runs the hardware
    mfma_scale decode kernel on ZERO-COPY strided views of the
    preallocated cache; tail runs the bf16 kernel; merged partials match
    the dequantized-union oracle at fp8-class bars (Q/P quantize in the
    MX kernel). Outlier channel included — plain fp8 would NaN."""
    from tree_attention_torch_amd.quant import (dequantize_k_mx,
                                                dequantize_v_mx,
                                                quantize_k_mx,
                                                quantize_v_mx)

    torch.manual_seed(7)
    b, hkv, hq, d = 1, 2, 8, 128
    sess = DecodeSession(b, hkv, d, max_tokens=8192, device="cuda",
                         kv_dtype="mx", block=256)
    t_all = 6500
    ks = torch.randn(b, hkv, t_all, d, device="cuda")
    vs = torch.randn(b, hkv, t_all, d, device="cuda")
    ks[..., 13] *= 700.0
    assert torch.isnan(ks.to(torch.float8_e4m3fn).float()).any()
    sess.prefill(ks[:, :, : t_all - 3], vs[:, :, : t_all - 3])
    for t in range(t_all - 3, t_all):
        sess.append(ks[:, :, t : t + 1], vs[:, :, t : t + 1])
    assert sess.q_len == t_all // 64 * 64 and \
        sess.tail_len == t_all - sess.q_len
    q = torch.randn(b, hq, 1, d, device="cuda").bfloat16()
    out = sess.attend(q)
    assert torch.isfinite(out).all()
    qn = sess.q_len
    kd = dequantize_k_mx(*quantize_k_mx(ks[:, :, :qn].bfloat16().float()))
    vd = dequantize_v_mx(*quantize_v_mx(vs[:, :, :qn].bfloat16().float()))
    ku = torch.cat([kd, ks[:, :, qn:].bfloat16().float()], dim=2).cpu()
    vu = torch.cat([vd, vs[:, :, qn:].bfloat16().float()], dim=2).cpu()
    # Q e4m3-simulated for the prefix; bf16 for the tail — use the softer
    # of the two (fp8-class) bars against the plain-q oracle
    ref, _ = flash_res_lse(q.float().cpu(), ku, vu)
    vmax = vu.abs().amax().clamp(min=1.0)
    assert ((out.cpu() - ref).abs() / vmax).max().item() < 4e-2


def test_session_block_cyclic_ownership_property():
    """Hypothesis property (pure accounting): for any world size, block
    size and append sequence, each rank's local_len equals the number of
    global tokens whose block-cyclic owner is that rank, and the mx
    quantized-prefix/tail split always satisfies q_len % 64 == 0,
    tail_len < 64, q_len + tail_len == local_len."""
    from hypothesis import given, settings, strategies as st

    @settings(max_examples=30, deadline=None)
    @given(st.integers(1, 8), st.sampled_from([64, 128, 256]),
           st.lists(st.integers(1, 200), min_size=1, max_size=5))
    def run(world, block, chunks):
        total = sum(chunks)
        sessions = [
            DecodeSession(1, 1, 128, max_tokens=max(total + block, 1024),
                          device="cpu", kv_dtype="mx", block=block)
            for _ in range(world)
        ]
        # monkey-set rank/world (no process group on CPU single-proc)
        for r, s in enumerate(sessions):
            s.rank, s.world = r, world
        ks = torch.randn(1, 1, total, 128)
        t = 0
        for n in chunks:
            for s in sessions:
                s.prefill(ks[:, :, t : t + n], ks[:, :, t : t + n])
            t += n
        for r, s in enumerate(sessions):
            expect = sum(1 for g in range(total)
                         if (g // block) % world == r)
            assert s.local_len == expect, (r, s.local_len, expect)
            assert s.q_len % 64 == 0 and 0 <= s.tail_len < 64
            assert s.q_len + s.tail_len == s.local_len
            assert s.total == total

    run()
