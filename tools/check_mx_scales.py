"""Pin the mfma_scale per-lane E8M0 scale-operand semantics on silicon:
each lane's fragment is one 32-elem k-block, its scale byte goes in bits
0..7 of the scale operand (byte_sel 0). Oracle applies 2^(s-127) per
block. Exact match required (fp8 products in fp32 accumulate)."""
import sys

sys.path.insert(0, "/root/repo")
import torch  # noqa: E402

from tree_attention_torch_amd.ops import flash  # noqa: E402

ext = flash._load_extension()
torch.manual_seed(0)
a = (torch.randn(32, 64, device="cuda") * 2).to(torch.float8_e4m3fn)
b = (torch.randn(64, 32, device="cuda") * 2).to(torch.float8_e4m3fn)
sa = torch.randint(120, 135, (32, 2), device="cuda", dtype=torch.uint8)
sb = torch.randint(120, 135, (2, 32), device="cuda", dtype=torch.uint8)
c = ext.probe_mfma_mx_scaled(a, b, sa, sb)

af = a.float()
bf = b.float()
fa = torch.pow(2.0, sa.float() - 127)        # (32 rows, 2 blocks)
fb = torch.pow(2.0, sb.float() - 127)        # (2 blocks, 32 cols)
ref = torch.zeros(32, 32, device="cuda")
for blk in range(2):
    asc = af[:, blk * 32:(blk + 1) * 32] * fa[:, blk:blk + 1]
    bsc = bf[blk * 32:(blk + 1) * 32, :] * fb[blk:blk + 1, :]
    ref += asc @ bsc
err = (c - ref).abs().max().item()
rel = err / ref.abs().max().item()
print(f"max|err|={err:.3e} rel={rel:.2e}")
assert rel < 1e-5, "scale semantics mismatch"
print("MX_SCALED_OK")
