import sys
sys.path.insert(0, "/root/repo")
import torch
from tree_attention_torch_amd.ops import flash
ext = flash._load_extension()
torch.manual_seed(1)
a = (torch.randn(32, 64, device="cuda") * 2).to(torch.float8_e4m3fn)
b = (torch.randn(64, 32, device="cuda") * 2).to(torch.float8_e4m3fn)
u127 = torch.full((32, 2), 127, device="cuda", dtype=torch.uint8)
u127b = torch.full((2, 32), 127, device="cuda", dtype=torch.uint8)
ref0 = a.float() @ b.float()
c0 = ext.probe_mfma_mx_scaled(a, b, u127, u127b)
e0 = (c0 - ref0).abs().max().item()
print(f"random data, unit scale tensors: max|err| = {e0:.3e} "
      f"(ref max {ref0.abs().max().item():.1f})")
# one scaled row on random data
sa = u127.clone(); sa[3, 0] = 128
c1 = ext.probe_mfma_mx_scaled(a, b, sa, u127b)
ref1 = ref0.clone()
ref1[3] = 2.0 * (a.float()[3, :32] @ b.float()[:32]) + \
    (a.float()[3, 32:] @ b.float()[32:])
e1 = (c1 - ref1).abs().max().item()
print(f"one scaled row, random data: max|err| = {e1:.3e}")
print("row3:", c1[3, :4].tolist(), "vs", ref1[3, :4].tolist())
