"""Measure actual fp8 kernel error vs the dequantized-KV fp32 oracle, to
set honest test tolerances."""
import sys

sys.path.insert(0, "/root/repo")
import torch  # noqa: E402

from tree_attention_torch_amd.ops import flash  # noqa: E402
from tree_attention_torch_amd.ops.reference import flash_res_lse  # noqa: E402

ext = flash._load_extension()
worst_o = worst_l = 0.0
for seed in range(5):
    torch.manual_seed(seed)
    for (b, hq, hkv, t, tq, causal) in [
            (1, 8, 8, 4096, 1, False), (1, 32, 4, 8192, 1, False),
            (2, 8, 2, 1500, 1, False), (1, 8, 8, 1024, 256, True),
            (1, 16, 2, 2048, 512, True)]:
        q = torch.randn(b, hq, tq, 128, device="cuda").bfloat16()
        k8 = torch.randn(b, hkv, t, 128, device="cuda").to(torch.float8_e4m3fn)
        v8 = torch.randn(b, hkv, t, 128, device="cuda").to(torch.float8_e4m3fn)
        scale = 128 ** -0.5
        o, l = ext.flash_attention(q, k8, v8, scale, causal, t - tq, 0)
        ro, rl = flash_res_lse(q.float().cpu(), k8.float().cpu(),
                               v8.float().cpu(), scale, causal, t - tq, 0)
        eo = (o.cpu() - ro).abs().max().item()
        el = (l.cpu() - rl).abs().max().item()
        worst_o = max(worst_o, eo)
        worst_l = max(worst_l, el)
        print(f"seed{seed} b{b} hq{hq} hkv{hkv} t{t} tq{tq} c{int(causal)}: "
              f"max|dO|={eo:.4f} max|dLSE|={el:.4f}", flush=True)
print(f"WORST: dO={worst_o:.4f} dLSE={worst_l:.4f}")
