import torch, time, sys
sys.path.insert(0, '/root/repo')
from tree_attention_torch_amd.ops.flash import local_attention
torch.manual_seed(0)
for (tq, tkv, causal) in [(8192, 8192, True), (8192, 8192, False), (4096, 131072, True)]:
    q = torch.randn(1, 32, tq, 128, device='cuda').bfloat16()
    k = torch.randn(1, 32, tkv, 128, device='cuda').bfloat16()
    v = torch.randn(1, 32, tkv, 128, device='cuda').bfloat16()
    for _ in range(3): local_attention(q, k, v, is_causal=causal, q_offset=tkv-tq)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    N = 10
    for _ in range(N): local_attention(q, k, v, is_causal=causal, q_offset=tkv-tq)
    torch.cuda.synchronize(); dt = (time.perf_counter()-t0)/N
    # valid (unmasked) score pairs: causal with q at end
    if causal:
        pairs = tq*(tkv-tq) + tq*(tq+1)//2
    else:
        pairs = tq*tkv
    flops = 2*2*1*32*pairs*128
    print(f"tq={tq} tkv={tkv} causal={causal}: {dt*1e3:.2f} ms  {flops/dt/1e12:.1f} TF/s")
