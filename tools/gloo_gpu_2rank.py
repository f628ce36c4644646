"""Two ranks sharing one GPU over gloo (CUDA tensors, CPU-bounced
collectives): exercises the full multi-rank GPU code path — HIP kernels +
sharded data + cross-rank combine — where RCCL refuses duplicate devices."""
import os
import sys

import torch
import torch.multiprocessing as mp


def worker(rank):
    sys.path.insert(0, "/root/repo")
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT="29722",
                      RANK=str(rank), WORLD_SIZE="2", LOCAL_RANK="0",
                      TREE_ATTN_BACKEND="gloo")
    from tree_attention_torch_amd.ops.reference import flash_res_lse
    from tree_attention_torch_amd.parallel.pg import cleanup, setup
    from tree_attention_torch_amd.parallel.tree import tree_attention

    setup(rank, 2)
    try:
        torch.manual_seed(0)
        tl = 4096
        q = torch.randn(1, 8, 1, 128).bfloat16().cuda()
        kf = torch.randn(1, 8, 2 * tl, 128).bfloat16().cuda()
        vf = torch.randn(1, 8, 2 * tl, 128).bfloat16().cuda()
        k = kf[:, :, rank * tl:(rank + 1) * tl].contiguous()
        v = vf[:, :, rank * tl:(rank + 1) * tl].contiguous()
        for strat in ("allgather", "allreduce"):
            out = tree_attention(q, k, v, combine=strat)
            ref, _ = flash_res_lse(q.cpu(), kf.cpu(), vf.cpu())
            torch.testing.assert_close(out.cpu(), ref, rtol=2.5e-2, atol=2.5e-2)
            if rank == 0:
                print(f"2-rank shared-GPU {strat}: OK", flush=True)
        # causal chunked prefill across the shards too
        tq = 512
        qp = torch.randn(1, 4, tq, 128).bfloat16().cuda()
        kp = torch.randn(1, 4, 2 * 1024, 128).bfloat16().cuda()
        vp = torch.randn(1, 4, 2 * 1024, 128).bfloat16().cuda()
        kl = kp[:, :, rank * 1024:(rank + 1) * 1024].contiguous()
        vl = vp[:, :, rank * 1024:(rank + 1) * 1024].contiguous()
        outp = tree_attention(qp, kl, vl, is_causal=True, q_chunk=256)
        refp, _ = flash_res_lse(qp.cpu(), kp.cpu(), vp.cpu(), is_causal=True,
                                q_offset=2 * 1024 - tq)
        torch.testing.assert_close(outp.cpu(), refp, rtol=3e-2, atol=3e-2)
        if rank == 0:
            print("2-rank shared-GPU causal chunked prefill: OK", flush=True)
    finally:
        cleanup()


if __name__ == "__main__":
    ctx = mp.get_context("spawn")
    ps = [ctx.Process(target=worker, args=(r,)) for r in range(2)]
    [p.start() for p in ps]
    [p.join(300) for p in ps]
    assert all(p.exitcode == 0 for p in ps), [p.exitcode for p in ps]
    print("2-rank-shared-GPU end-to-end: PASS")
