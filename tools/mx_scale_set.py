"""Identify exactly WHICH 32 k-elements a lane's scale byte applies to:
b's columns are bit indicators of k, so C[3][j] = 32 + |{k in S: bit_j(k)}|
where S is the scaled set (w=2 on S). Column 6 = all ones -> |S| check."""
import sys
sys.path.insert(0, "/root/repo")
import torch
from tree_attention_torch_amd.ops import flash
ext = flash._load_extension()
a = torch.full((32, 64), 1.0, device="cuda").to(torch.float8_e4m3fn)
bm = torch.zeros(64, 32, device="cuda")
for j in range(6):
    for k in range(64):
        bm[k, j] = float((k >> j) & 1)
bm[:, 6] = 1.0
b = bm.to(torch.float8_e4m3fn)
u127 = torch.full((32, 2), 127, device="cuda", dtype=torch.uint8)
u127b = torch.full((2, 32), 127, device="cuda", dtype=torch.uint8)
base = ext.probe_mfma_mx_scaled(a, b, u127, u127b)
# scale byte sourced from LANE 3 (current kernel: lane3 reads sa[3][1])
for (r, blk, lane) in [(3, 1, "lane3"), (3, 0, "lane35")]:
    sa = u127.clone(); sa[r, blk] = 128
    c = ext.probe_mfma_mx_scaled(a, b, sa, u127b)
    extra = [int(round(c[3, j].item() - base[3, j].item())) for j in range(7)]
    print(f"scale from {lane}: extra per bit [b0..b5, all] = {extra}")
    # reconstruct S assuming S = {k: matches}; print which k-bits are 'all in'
