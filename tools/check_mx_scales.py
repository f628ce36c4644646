"""Pin the mfma_scale_f32_32x32x64_f8f6f4 per-block E8M0 scale semantics
on silicon (the basis of the MX-scaled fp8 path).

Measured semantics (this tool + tools/mx_scale_set.py, MI355X 2026-09-14):
  * the E8M0 byte decodes as 2^(s-127), exact across 119..134;
  * lane l's scale byte (bits 0..7 of the scale operand, sel 0) applies
    to scale block (l>>5) of A-row (l&31) / B-col (l&31);
  * a scale BLOCK is the INTERLEAVED logical k-set {k: (k>>4)&1 == blk}
    — the instruction's internal k-order swaps bits 4 and 5 of the
    logical contraction index, so the "32 consecutive internal k" block
    is {0-15, 32-47} / {16-31, 48-63} in logical k. All-ones data cannot
    see this (both halves weigh equally); bit-indicator columns can.
"""
import sys

sys.path.insert(0, "/root/repo")
import torch  # noqa: E402

from tree_attention_torch_amd.ops import flash  # noqa: E402


def mx_oracle(a, b, sa, sb):
    """Per-block-scaled matmul with the measured interleaved blocks."""
    af, bf = a.float(), b.float()
    fa = torch.pow(2.0, sa.float() - 127)   # (32 rows, 2 blocks)
    fb = torch.pow(2.0, sb.float() - 127)   # (2 blocks, 32 cols)
    k = torch.arange(64, device=a.device)
    ref = torch.zeros(32, 32, device=a.device)
    mag = torch.zeros(32, 32, device=a.device)
    for blk in range(2):
        sel = ((k >> 4) & 1) == blk
        asc = af[:, sel] * fa[:, blk:blk + 1]
        bsc = bf[sel, :] * fb[blk:blk + 1, :]
        ref += asc @ bsc
        mag += asc.abs() @ bsc.abs()
    return ref, mag


def run(name, seed, lo, hi):
    torch.manual_seed(seed)
    ext = flash._load_extension()
    a = (torch.randn(32, 64, device="cuda") * 2).to(torch.float8_e4m3fn)
    b = (torch.randn(64, 32, device="cuda") * 2).to(torch.float8_e4m3fn)
    sa = torch.randint(lo, hi, (32, 2), device="cuda", dtype=torch.uint8)
    sb = torch.randint(lo, hi, (2, 32), device="cuda", dtype=torch.uint8)
    c = ext.probe_mfma_mx_scaled(a, b, sa, sb)
    ref, mag = mx_oracle(a, b, sa, sb)
    rel = ((c - ref).abs() / mag.clamp(min=1e-30)).max().item()
    print(f"{name}: rel-to-magnitude {rel:.2e}")
    return rel


if __name__ == "__main__":
    worst = max(run("narrow-scales", 1, 125, 130),
                run("wide-scales", 2, 120, 135),
                run("unit", 3, 127, 128))
    assert worst < 2e-4, f"scale semantics mismatch: {worst}"
    print("MX_SCALED_OK")
