import sys
sys.path.insert(0, "/root/repo")
import torch
from tree_attention_torch_amd.ops.flash import local_attention_mx
from tree_attention_torch_amd.ops.reference import flash_res_lse
from tree_attention_torch_amd.quant import *

def case(ch):
    torch.manual_seed(7)
    b, hq, hkv, t, tq = 1, 1, 1, 128, 1
    q = torch.randn(b, hq, tq, 128, device="cuda").bfloat16()
    k = torch.randn(b, hkv, t, 128, device="cuda")
    v = torch.randn(b, hkv, t, 128, device="cuda")
    if ch >= 0: k[..., ch] *= 700.0
    k8, ks = quantize_k_mx(k); v8, vs = quantize_v_mx(v)
    out, lse = local_attention_mx(q, k8, ks, v8, vs, is_causal=False, q_offset=t-tq)
    q_sim = q.float().to(torch.float8_e4m3fn).float()
    ro, rl = flash_res_lse(q_sim.cpu(), dequantize_k_mx(k8,ks).cpu(),
                           dequantize_v_mx(v8,vs).cpu(), is_causal=False, q_offset=t-tq)
    g = (ch >> 6) * 2 + ((ch >> 4) & 1) if ch >= 0 else -1
    print(f"ch={ch:4d} (group {g}): lse_err={(lse.cpu()-rl).abs().max().item():.4f} "
          f"sample ks={ks[0,0,0].tolist()}")

for ch in (-1, 13, 20, 40, 52, 77, 90, 110, 125):
    case(ch)
