"""Decode effective-bandwidth check: bf16 vs fp8 MHA/GQA at 128K."""
import sys, time
sys.path.insert(0, "/root/repo")
import torch
from tree_attention_torch_amd.ops.flash import local_attention

for dtype, hkv in (("bf16", 32), ("fp8", 32), ("bf16", 4), ("fp8", 4)):
    torch.manual_seed(0)
    t = 131072
    q = torch.randn(1, 32, 1, 128, device="cuda").bfloat16()
    if dtype == "fp8":
        k = torch.randn(1, hkv, t, 128, device="cuda").to(torch.float8_e4m3fn)
    else:
        k = torch.randn(1, hkv, t, 128, device="cuda").bfloat16()
    v = k.clone()
    for _ in range(5):
        local_attention(q, k, v)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(50):
        local_attention(q, k, v)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 50
    byt = 2 * hkv * t * 128 * (1 if dtype == "fp8" else 2)
    print(f"{dtype} hkv={hkv} 128K: {dt*1e3:.3f} ms  {byt/dt/1e12:.2f} TB/s")
