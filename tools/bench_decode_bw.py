"""Decode effective-bandwidth check: bf16 vs fp8 vs MX-fp8 MHA/GQA at 128K."""
import sys, time
sys.path.insert(0, "/root/repo")
import torch
from tree_attention_torch_amd.ops.flash import local_attention, local_attention_mx
from tree_attention_torch_amd.quant import quantize_k_mx, quantize_v_mx

for dtype, hkv in (("bf16", 32), ("fp8", 32), ("mx", 32),
                   ("bf16", 4), ("fp8", 4), ("mx", 4)):
    torch.manual_seed(0)
    t = 131072
    q = torch.randn(1, 32, 1, 128, device="cuda").bfloat16()
    if dtype == "fp8":
        k = torch.randn(1, hkv, t, 128, device="cuda").to(torch.float8_e4m3fn)
        v = k.clone()
        run = lambda: local_attention(q, k, v)
    elif dtype == "mx":
        kf = torch.randn(1, hkv, t, 128, device="cuda")
        k, ks = quantize_k_mx(kf)
        v, vs = quantize_v_mx(kf)
        del kf
        run = lambda: local_attention_mx(q, k, ks, v, vs)
    else:
        k = torch.randn(1, hkv, t, 128, device="cuda").bfloat16()
        v = k.clone()
        run = lambda: local_attention(q, k, v)
    for _ in range(5):
        run()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(50):
        run()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 50
    elem = 1 if dtype in ("fp8", "mx") else 2
    byt = 2 * hkv * t * 128 * elem
    if dtype == "mx":  # + K scales (T*4 B) + V scales (T/32*128 B) per head
        byt += 2 * hkv * t * 4
    print(f"{dtype} hkv={hkv} 128K: {dt*1e3:.3f} ms  {byt/dt/1e12:.2f} TB/s")
