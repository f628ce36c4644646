"""Unit checks for the generated gen6 asm streams vs torch references."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from tree_attention_torch_amd.ops import flash

ext = flash._load_extension()
torch.manual_seed(0)
q = (torch.randn(64, 128) * 0.5).bfloat16().cuda()
k = (torch.randn(32, 128) * 0.5).bfloat16().cuda()
out = ext.probe_gen6_qkt(q, k).cpu()  # (2, 64, 16)
ref = (k.float() @ q.float().T).cpu()  # (32 keys, 64 rows)
lanes = torch.arange(64)
row32 = lanes % 32
h = lanes // 32
ok = True
for j in range(2):
    got = out[j]  # (64, 16)
    for reg in range(16):
        key = (reg % 4) + 8 * (reg // 4)
        # per lane: key + 4*h, qrow = j*32 + row32
        exp = ref[(key + 4 * h), (j * 32 + row32)]
        err = (got[:, reg] - exp).abs().max().item()
        if err > 0.05:
            ok = False
            print(f"j={j} reg={reg}: maxerr {err:.4f}")
print("GEN6 QKT:", "OK" if ok else "FAIL",
      " overall maxerr:", float((out[0] - torch.stack(
          [ref[((torch.arange(16) % 4) + 8 * (torch.arange(16) // 4))[r] + 4 * h,
               0 * 32 + row32] for r in range(16)], dim=1)).abs().max()))
