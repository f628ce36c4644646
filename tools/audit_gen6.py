#!/usr/bin/env python3
"""Audit the gen6 kernel's .s: the O accumulators (literal a[0:127]) carry
hand-owned state ACROSS asm statements in the main tile loop, so a COMPILER
write into a0..a127 inside that loop is silent corruption (guide §5.7 item
4). Stream scratch v[144:255] and s[40:41] live only WITHIN one statement
(clobbered there) — compiler use of those between statements is harmless.

Method: in each fa_prefill6_kernel section, reconstruct the asm main loop
from the branch graph (backedge targets/sources bracket the loop body;
layout order is not execution order, so a linear source region is wrong).
Then flag any line OUTSIDE ;;#ASMSTART/;;#ASMEND inside the loop span that
writes a0..a127, and any branch from outside the span back into it (which
would put unaudited code on the loop path). Also reports spill counts.

Usage: python tools/audit_gen6.py <fa_kernels .s>
"""

import re
import sys


def audit(path: str) -> int:
    text = open(path).read()
    errors = 0
    secs = re.split(r"\n(?=[.\w$]*_ZN2ta18fa_prefill6_kernel)", text)
    kernels = [s for s in secs if s.startswith("_ZN2ta18fa_prefill6_kernel")]
    if not kernels:
        print("AUDIT: no fa_prefill6_kernel section found")
        return 1

    for sec in kernels:
        lines = sec.splitlines()
        name = lines[0].rstrip(":")[:60]
        labels = {}
        for i, ln in enumerate(lines):
            m = re.match(r"^(\.?[\w$.]+):", ln)
            if m:
                labels[m.group(1)] = i
        branches = []
        for i, ln in enumerate(lines):
            m = re.search(r"s_(?:branch|cbranch\w*)\s+(\.?[\w$.]+)", ln)
            if m and m.group(1) in labels:
                branches.append((i, labels[m.group(1)]))
        # the main asm loop is the backedge group whose span contains the
        # tile statement — identified by its RSFN (shared rescale) label
        rsfn = None
        for i, ln in enumerate(lines):
            if "RSFN" in ln and ln.rstrip().endswith(":"):
                rsfn = i
                break
        if rsfn is None:
            print(f"AUDIT {name}: RSFN marker not found")
            errors += 1
            continue
        spans = [(t, i) for i, t in branches if t < i and t <= rsfn <= i]
        if not spans:
            print(f"AUDIT {name}: no main-loop backedge around the tile "
                  "statement; FAIL")
            errors += 1
            continue
        lo = min(t for t, _ in spans)
        hi = max(i for _, i in spans)
        # external jumps INTO the loop span (other than entry edges from
        # the immediately preceding pre-header block)
        external = [(i, t) for i, t in branches
                    if (i < lo - 16 or i > hi) and lo < t < hi]
        if external:
            errors += 1
            print(f"AUDIT {name}: branches into the loop span from outside: "
                  f"{external[:5]}")
        in_asm = False
        bad = []
        for i, ln in enumerate(lines):
            if ";;#ASMSTART" in ln:
                in_asm = True
                continue
            if ";;#ASMEND" in ln:
                in_asm = False
                continue
            if in_asm or not (lo <= i <= hi):
                continue
            m = re.match(r"\s*([a-z0-9_]+)\s+a\[?(\d+)", ln)
            if m and not m.group(1).startswith("s_") and \
                    "read" not in m.group(1):
                n = int(m.group(2))
                if n < 128:
                    bad.append((i, ln.strip()))
        if bad:
            errors += 1
            print(f"AUDIT {name}: {len(bad)} compiler writes into a0..a127 "
                  f"inside the loop span [{lo}..{hi}]:")
            for i, ln in bad[:10]:
                print(f"  line {i}: {ln}")
        else:
            print(f"AUDIT {name}: clean (loop span [{lo}..{hi}])")

    spills = [int(x) for x in re.findall(r"\.vgpr_spill_count:\s*(\d+)", text)]
    names = re.findall(r"\.name:\s+(\S+)", text)
    for n, s in zip(names, spills):
        if "prefill6" in n and s:
            print(f"AUDIT: prefill6 kernel {n} has vgpr_spill_count={s}")
            errors += 1
    print("AUDIT:", "FAIL" if errors else "PASS")
    return 1 if errors else 0


if __name__ == "__main__":
    sys.exit(audit(sys.argv[1] if len(sys.argv) > 1 else
                   "fa_kernels-hip-amdgcn-amd-amdhsa-gfx950.s"))
