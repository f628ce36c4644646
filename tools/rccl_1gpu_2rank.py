"""Sanity: the RCCL tree-combine path with 2 ranks sharing one GPU.

NOTE: RCCL refuses two ranks on one device ("Duplicate GPU detected"), so
this cannot pass on a 1-GPU box — kept as the ready-made check for any
multi-GPU box. The collective logic itself is covered by the gloo
world_size 2/4 tests in tests/test_sharded_cpu.py."""
import os, sys, torch
import torch.multiprocessing as mp

def worker(rank):
    sys.path.insert(0, '/root/repo')
    import torch.distributed as dist
    os.environ.update(MASTER_ADDR='127.0.0.1', MASTER_PORT='29571',
                      RANK=str(rank), WORLD_SIZE='2', LOCAL_RANK='0')
    torch.cuda.set_device(0)
    dist.init_process_group('nccl', rank=rank, world_size=2)
    try:
        from tree_attention_torch_amd.parallel.tree import tree_attention
        from tree_attention_torch_amd.ops.reference import flash_res_lse
        torch.manual_seed(0)
        t_total, tl = 8192, 4096
        q = torch.randn(1, 8, 1, 128).bfloat16().cuda()
        k_full = torch.randn(1, 8, t_total, 128).bfloat16().cuda()
        v_full = torch.randn(1, 8, t_total, 128).bfloat16().cuda()
        k = k_full[..., rank*tl:(rank+1)*tl, :].contiguous()
        v = v_full[..., rank*tl:(rank+1)*tl, :].contiguous()
        for strat in ('allgather', 'allreduce'):
            out = tree_attention(q, k, v, combine=strat)
            ref, _ = flash_res_lse(q.cpu(), k_full.cpu(), v_full.cpu())
            torch.testing.assert_close(out.cpu(), ref, rtol=2.5e-2, atol=2.5e-2)
            if rank == 0: print(f"rccl 2-rank {strat}: OK")
    finally:
        dist.destroy_process_group()

if __name__ == '__main__':
    ctx = mp.get_context('spawn')
    ps = [ctx.Process(target=worker, args=(r,)) for r in range(2)]
    [p.start() for p in ps]
    [p.join(120) for p in ps]
    assert all(p.exitcode == 0 for p in ps), [p.exitcode for p in ps]
    print("RCCL 2-rank-1-GPU: PASS")
