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

# ---- softmax+pack stream unit ----
torch.manual_seed(2)
s_in = (torch.randn(64, 16) * 3).float().cuda()
ml = torch.stack([torch.randn(64) * 2 - 1, torch.rand(64) * 5 + 0.1], dim=1).cuda()
# make rows share state across the half-wave (lane l and l+32 carry the same
# q-row in the real kernel); not required for the unit, it just mirrors use.
cl2 = 0.125 * 1.44269504
c_out, ml_out = [t.cpu() for t in ext.probe_gen6_softmax(s_in, ml, cl2)]
dc = (c_out[0] != c_out[1]).sum().item()
dml = (ml_out[0] - ml_out[1]).abs().max().item()
# unpack c as bf16 pairs and compare numerically too (rounding-path slack)
def unpack(c):
    lo = (c & 0xffff).to(torch.int32).to(torch.uint16).view(torch.bfloat16).float()
    hi = (c >> 16).to(torch.int32).to(torch.uint16).view(torch.bfloat16).float()
    return lo, hi
a_lo, a_hi = unpack(c_out[0].to(torch.int64))
b_lo, b_hi = unpack(c_out[1].to(torch.int64))
dnum = max((a_lo - b_lo).abs().max().item(), (a_hi - b_hi).abs().max().item())
print(f"GEN6 SOFTMAX: c bit-mismatches={dc} c value-diff={dnum:.6f} "
      f"m/l/alpha maxdiff={dml:.2e}",
      "OK" if (dml < 1e-5 and dnum < 1e-2) else "FAIL")
mm = (c_out[0] != c_out[1])  # (64, 8)
print("mismatch by c-index:", mm.sum(dim=0).tolist())
print("mismatch by half-wave: lo", mm[:32].sum().item(), "hi", mm[32:].sum().item())
l0 = 0
print("lane0 asm c:", [hex(x) for x in c_out[0][l0].tolist()])
print("lane0 C   c:", [hex(x) for x in c_out[1][l0].tolist()])
l0 = 40
print("lane40 asm c:", [hex(x) for x in c_out[0][l0].tolist()])
print("lane40 C   c:", [hex(x) for x in c_out[1][l0].tolist()])
