import sys
sys.path.insert(0, "/root/repo")
import torch
from tree_attention_torch_amd.ops import flash
ext = flash._load_extension()
a = torch.full((32, 64), 1.0, device="cuda").to(torch.float8_e4m3fn)
b = torch.full((64, 32), 1.0, device="cuda").to(torch.float8_e4m3fn)
u127b = torch.full((2, 32), 127, device="cuda", dtype=torch.uint8)
torch.manual_seed(0)
sa = torch.randint(125, 130, (32, 2), device="cuda", dtype=torch.uint8)
c = ext.probe_mfma_mx_scaled(a, b, sa, u127b)
fa = torch.pow(2.0, sa.float() - 127)
exp = 32.0 * (fa[:, 0] + fa[:, 1])
print("row: sa0 sa1 measured expected")
ok = True
for r in range(8):
    m = c[r, 0].item(); e = exp[r].item()
    ok &= abs(m - e) < 1e-3 * max(e, 1)
    print(f"{r}: {sa[r,0].item()} {sa[r,1].item()}  {m:.3f}  {e:.3f}")
print("cols uniform?", bool((c.max(dim=1).values == c.min(dim=1).values).all()))
print("ALL_ROWS_MATCH" if ok else "MISMATCH")
# two deviations in the SAME lane-pair vs different rows
for pairs in ([(0,0,128)], [(0,0,128),(0,1,129)], [(0,0,128),(1,0,129)],
              [(0,0,128),(17,1,125)]):
    sa2 = torch.full((32, 2), 127, device="cuda", dtype=torch.uint8)
    for r, blk, v in pairs:
        sa2[r, blk] = v
    c2 = ext.probe_mfma_mx_scaled(a, b, sa2, u127b)
    fa2 = torch.pow(2.0, sa2.float() - 127)
    exp2 = 32.0 * (fa2[:, 0] + fa2[:, 1])
    rows = [r for r, _, _ in pairs]
    print(pairs, "->", [(r, round(c2[r,0].item(),3), round(exp2[r].item(),3))
                        for r in sorted(set(rows))])
