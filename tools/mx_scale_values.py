import sys
sys.path.insert(0, "/root/repo")
import torch
from tree_attention_torch_amd.ops import flash
ext = flash._load_extension()
a = torch.full((32, 64), 1.0, device="cuda").to(torch.float8_e4m3fn)
b = torch.full((64, 32), 1.0, device="cuda").to(torch.float8_e4m3fn)
u127 = torch.full((32, 2), 127, device="cuda", dtype=torch.uint8)
u127b = torch.full((2, 32), 127, device="cuda", dtype=torch.uint8)
base = ext.probe_mfma_mx_scaled(a, b, u127, u127b)[0, 0].item()
print("base:", base)
for sval in (119, 124, 125, 126, 127, 128, 129, 130, 134):
    sa = u127.clone(); sa[0, 0] = sval
    c = ext.probe_mfma_mx_scaled(a, b, sa, u127b)[0, 0].item()
    blk = (c - 32.0) / 32.0   # block-0 factor (block 1 stays x1)
    print(f"s={sval}: C00={c:.6f} block0_factor={blk:.6f} "
          f"expected 2^{sval-127}={2.0**(sval-127):.6f}")
