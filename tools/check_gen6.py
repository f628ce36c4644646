"""Unit checks for the generated gen6 asm streams vs torch references."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from tree_attention_torch_amd.ops import flash

ext = flash._load_extension()
torch.manual_seed(0)
q = (torch.randn(64, 128) * 0.5).bfloat16().cuda()
k = (torch.randn(32, 128) * 0.5).bfloat16().cuda()
out = ext.probe_gen6_qkt(q, k).cpu()  # (4, 64, 16): asm j0/j1, builtin j0/j1
ref = (k.float() @ q.float().T).cpu()  # (32 keys, 64 rows)
lanes = torch.arange(64)
row32 = lanes % 32
h = lanes // 32
def against_ref(got, j):
    bad = 0
    mx = 0.0
    for reg in range(16):
        key = (reg % 4) + 8 * (reg // 4)
        exp = ref[(key + 4 * h), (j * 32 + row32)]
        err = (got[:, reg] - exp).abs().max().item()
        mx = max(mx, err)
        if err > 0.05:
            bad += 1
    return bad, mx

for j in range(2):
    ba, ma = against_ref(out[j], j)       # asm vs torch-ref
    bc, mc = against_ref(out[2 + j], j)   # builtin vs torch-ref
    av = (out[j] - out[2 + j]).abs().max().item()  # asm vs builtin
    print(f"j={j}: asm-vs-ref bad={ba} maxerr={ma:.4f} | "
          f"builtin-vs-ref bad={bc} maxerr={mc:.4f} | asm-vs-builtin {av:.4f}")
