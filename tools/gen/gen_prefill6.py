#!/usr/bin/env python3
"""Generate the hand-placed asm streams for fa_prefill6 (gfx950).

Emits tree_attention_torch_amd/ops/hip/fa_prefill6_gen.h, included by
fa_kernels.hip. The prefill6 kernel is the 1-wave/SIMD hand-scheduled
regime (the guide's pwg4x64 direction): 4 waves per 256-row q-block, each
wave two 32-row q-blocks, whole 512-register file per lane. The per-TILE
main loop body is ONE asm volatile statement so no hand-owned register
state ever crosses a statement boundary the compiler could interleave
with; O accumulators live in literal AGPRs a[0:127] (clobbered), Q
fragments are operand-passed AGPRs, scores/packs/temps are clobbered
fixed arch VGPRs v[144:255], and m/l state crosses tiles as "+v"
operands. C code around the statements does staging (glds, counted
vmcnt), addressing, the boundary-tile fallback and the epilogue.

Verified silicon rules baked in (docs/ROUND1_NOTES.md):
  * MFMA D -> VALU reader: 12 wait states (s_nop 11 at stream seams);
    D-as-C accumulate chains need none; never pad inside a chain.
  * VGPR write -> v_permlane* reader: s_nop 1.
  * v_exp_f32 (TRANS) result -> reader: one wait state.
  * ds_read in-asm with counted lgkmcnt (one-ahead pipelining).
  * The XOR swizzles are nb-independent: K/V slice addresses are
    lane-resolved registers + an immediate ds offset per (nb, ks).
  * Multi-instruction outputs need "=&v" earlyclobber.

Register map (fixed, all clobbered by the tile statement):
  a[0:127]    O accumulators: a[j*64 + nd*16 + r]
  v[192:207]  sjA0   v[208:223] sjA1   (scores, buffer A)
  v[224:239]  sjB0   v[240:255] sjB1   (scores, buffer B — stagger)
  v[176:183]  c0 packs   v[184:191] c1 packs
  v[160:163]  kr0        v[164:167] kr1   (K read pipeline)
  v[144:155]  vr: three 4-reg V-fragment slots (tr-read pipeline)
  v[168:175]  temps: T0 T1 MT0 MT1 RS P0 P1 spare
  v[156:159]  AL0 AL1 spare spare
"""

import os

OUT = os.path.join(os.path.dirname(__file__), "..", "..",
                   "tree_attention_torch_amd", "ops", "hip",
                   "fa_prefill6_gen.h")

MFMA = "v_mfma_f32_32x32x16_bf16"

# ---- fixed register map ----
SJ = {("A", 0): 192, ("A", 1): 208, ("B", 0): 224, ("B", 1): 240}
CP = {0: 176, 1: 184}
KR = {0: 160, 1: 164}
VRSLOT = [144, 148, 152]          # three 4-reg V fragment slots
T0, T1 = "v168", "v169"
MT = {0: "v170", 1: "v171"}
RS, P0, P1 = "v172", "v173", "v174"
AL = {0: "v156", 1: "v157"}
# zipped-stream per-block temp sets (VERDICT: at 1 wave/SIMD nothing hides
# dependent-VALU latency, so the two blocks' softmax chains interleave at
# instruction granularity — consecutive ops independent). Block 1's p1/rs
# borrow v144/v145 from the VR pool: softmax fills run in QKT phases, the
# V-fragment slots only in PV phases — lifetimes disjoint.
TT = {0: ("v168", "v169"), 1: ("v158", "v159")}   # permlane temp pairs
PP = {0: ("v172", "v173"), 1: ("v175", "v144")}   # p0/p1 per block
RSJ = {0: "v174", 1: "v145"}                      # rowsum per block

# tile-statement operand indices
OP_M = {0: "%0", 1: "%1"}
OP_L = {0: "%2", 1: "%3"}
OP_KA = lambda s: f"%{4 + s}"          # noqa: E731
OP_VA = lambda c, nd: f"%{12 + c * 4 + nd}"   # noqa: E731
OP_CL2 = "%20"
# Q fragments live in literal AGPRs a[128:191], written once by the
# prologue LOAD_Q macros. Passing them as 16 "a" operands instead made
# the register allocator REMATERIALIZE the global loads inside every tile
# iteration, and hipcc brackets each reload with s_waitcnt vmcnt(0) —
# draining the glds staging pipeline 8x per tile (measured: gen6 at
# prefill2 parity instead of past it; guide §5 ".s-level trap (b)").
QBASE = 128


def OP_Q(j, s):
    b = QBASE + j * 32 + s * 4
    return f"a[{b}:{b + 3}]"


def vr(base, n=4):
    return f"v[{base}:{base + n - 1}]"


def sj_reg(buf, j, i):
    return f"v{SJ[(buf, j)] + i}"


# ---------------------------------------------------------------------------
# Instruction-stream builders. Each returns a list of (text, kind) where
# kind in {"mfma", "valu", "trans", "ds", "wait", "nop", "salu"} — the v2
# scheduler interleaves by kind; v1 just concatenates.
# ---------------------------------------------------------------------------

def qkt_stream(nb, buf):
    """QK^T for both q-blocks over 32-key block nb: 16 MFMAs, K slices via
    in-asm pipelined ds_read_b128 (one read ahead, counted lgkmcnt).
    Writes sj{buf}0 / sj{buf}1 (fixed v-ranges); A = kr pipeline; B = Q
    operand AGPRs. Probe-verified stream (probe_gen6_qkt)."""
    off = nb * 8192
    s0 = f"v[{SJ[(buf, 0)]}:{SJ[(buf, 0)] + 15}]"
    s1 = f"v[{SJ[(buf, 1)]}:{SJ[(buf, 1)] + 15}]"
    kr0, kr1 = vr(KR[0]), vr(KR[1])
    L = []
    L.append((f"ds_read_b128 {kr0}, {OP_KA(0)} offset:{off}", "ds"))
    L.append((f"ds_read_b128 {kr1}, {OP_KA(1)} offset:{off}", "ds"))
    L.append(("s_waitcnt lgkmcnt(1)", "wait"))
    L.append((f"{MFMA} {s0}, {kr0}, {OP_Q(0, 0)}, 0", "mfma"))
    L.append((f"{MFMA} {s1}, {kr0}, {OP_Q(1, 0)}, 0", "mfma"))
    for s in range(1, 8):
        cur = kr1 if (s % 2) else kr0
        nxt = kr0 if (s % 2) else kr1
        if s + 1 < 8:
            L.append((f"ds_read_b128 {nxt}, {OP_KA(s + 1)} offset:{off}", "ds"))
            L.append(("s_waitcnt lgkmcnt(1)", "wait"))
        else:
            L.append(("s_waitcnt lgkmcnt(0)", "wait"))
        L.append((f"{MFMA} {s0}, {cur}, {OP_Q(0, s)}, {s0}", "mfma"))
        L.append((f"{MFMA} {s1}, {cur}, {OP_Q(1, s)}, {s1}", "mfma"))
    return L


def sm_start_stream(j, buf):
    """startSM: in-lane + cross-half max of sj{buf}{j}, m_new (exp2 domain),
    alpha; updates m_run. MT{j} carries m_new to finishSM. Stream is the
    probe-verified SOFTMAX_PACK head with physical registers."""
    s = lambda i: sj_reg(buf, j, i)  # noqa: E731
    mt, al, m = MT[j], AL[j], OP_M[j]
    L = [
        (f"v_max3_f32 {mt}, {s(0)}, {s(1)}, {s(2)}", "valu"),
        (f"v_max3_f32 {T0}, {s(3)}, {s(4)}, {s(5)}", "valu"),
        (f"v_max3_f32 {T1}, {s(6)}, {s(7)}, {s(8)}", "valu"),
        (f"v_max3_f32 {mt}, {mt}, {T0}, {T1}", "valu"),
        (f"v_max3_f32 {T0}, {s(9)}, {s(10)}, {s(11)}", "valu"),
        (f"v_max3_f32 {T1}, {s(12)}, {s(13)}, {s(14)}", "valu"),
        (f"v_max3_f32 {mt}, {mt}, {T0}, {T1}", "valu"),
        (f"v_max3_f32 {mt}, {mt}, {s(15)}, {s(15)}", "valu"),
        (f"v_mov_b32 {T0}, {mt}", "valu"),
        (f"v_mov_b32 {T1}, {mt}", "valu"),
        ("s_nop 1", "nop"),  # VGPR write -> v_permlane hazard
        (f"v_permlane32_swap_b32 {T0}, {T1}", "valu"),
        (f"v_max_f32 {mt}, {T0}, {T1}", "valu"),
        (f"v_mul_f32 {mt}, {mt}, {OP_CL2}", "valu"),
        (f"v_max_f32 {mt}, {m}, {mt}", "valu"),          # m_new
        (f"v_sub_f32 {T0}, {m}, {mt}", "valu"),
        (f"v_exp_f32 {al}, {T0}", "trans"),              # alpha
        (f"v_mov_b32 {m}, {mt}", "valu"),                # m_run = m_new
    ]
    return L


def sm_finish_stream(j, buf):
    """finishSM: exp2+pack of sj{buf}{j} against MT{j}, row sum, l update,
    pswap of the c{j} packs. Probe-verified tail with physical registers."""
    s = lambda i: sj_reg(buf, j, i)  # noqa: E731
    mt, al, l = MT[j], AL[j], OP_L[j]
    c = lambda i: f"v{CP[j] + i}"  # noqa: E731
    L = [(f"v_mov_b32 {RS}, 0", "valu")]
    for i in range(8):
        L += [
            (f"v_fma_f32 {P0}, {s(2 * i)}, {OP_CL2}, -{mt}", "valu"),
            (f"v_fma_f32 {P1}, {s(2 * i + 1)}, {OP_CL2}, -{mt}", "valu"),
            (f"v_exp_f32 {P0}, {P0}", "trans"),
            (f"v_exp_f32 {P1}, {P1}", "trans"),
            ("s_nop 0", "nop"),  # TRANS result needs one state
            (f"v_cvt_pk_bf16_f32 {c(i)}, {P0}, {P1}", "valu"),
            (f"v_add_f32 {P0}, {P0}, {P1}", "valu"),
            (f"v_add_f32 {RS}, {RS}, {P0}", "valu"),
        ]
    L += [
        (f"v_mov_b32 {T0}, {RS}", "valu"),
        (f"v_mov_b32 {T1}, {RS}", "valu"),
        ("s_nop 1", "nop"),
        (f"v_permlane32_swap_b32 {T0}, {T1}", "valu"),
        (f"v_add_f32 {RS}, {T0}, {T1}", "valu"),
        (f"v_fma_f32 {l}, {l}, {al}, {RS}", "valu"),      # l = l*alpha + rs
        ("s_nop 1", "nop"),
        (f"v_permlane32_swap_b32 {c(0)}, {c(2)}", "valu"),
        (f"v_permlane32_swap_b32 {c(1)}, {c(3)}", "valu"),
        (f"v_permlane32_swap_b32 {c(4)}, {c(6)}", "valu"),
        (f"v_permlane32_swap_b32 {c(5)}, {c(7)}", "valu"),
    ]
    return L


def zip_streams(a, b):
    """Interleave two independent instruction streams one-for-one (nops
    and labels pass through without pairing)."""
    out = []
    ia = ib = 0
    while ia < len(a) or ib < len(b):
        if ia < len(a):
            out.append(a[ia])
            ia += 1
        if ib < len(b):
            out.append(b[ib])
            ib += 1
    return out


def _sm_start_one(j, buf):
    """One block's startSM with per-block temps (for zipping)."""
    s = lambda i: sj_reg(buf, j, i)  # noqa: E731
    t0, t1 = TT[j]
    mt, al, m = MT[j], AL[j], OP_M[j]
    return [
        (f"v_max3_f32 {mt}, {s(0)}, {s(1)}, {s(2)}", "valu"),
        (f"v_max3_f32 {t0}, {s(3)}, {s(4)}, {s(5)}", "valu"),
        (f"v_max3_f32 {t1}, {s(6)}, {s(7)}, {s(8)}", "valu"),
        (f"v_max3_f32 {mt}, {mt}, {t0}, {t1}", "valu"),
        (f"v_max3_f32 {t0}, {s(9)}, {s(10)}, {s(11)}", "valu"),
        (f"v_max3_f32 {t1}, {s(12)}, {s(13)}, {s(14)}", "valu"),
        (f"v_max3_f32 {mt}, {mt}, {t0}, {t1}", "valu"),
        (f"v_max3_f32 {mt}, {mt}, {s(15)}, {s(15)}", "valu"),
        (f"v_mov_b32 {t0}, {mt}", "valu"),
        (f"v_mov_b32 {t1}, {mt}", "valu"),
        ("s_nop 0", "nop"),  # VGPR write -> v_permlane (zip partner covers 1)
        (f"v_permlane32_swap_b32 {t0}, {t1}", "valu"),
        (f"v_max_f32 {mt}, {t0}, {t1}", "valu"),
        (f"v_mul_f32 {mt}, {mt}, {OP_CL2}", "valu"),
        (f"v_max_f32 {mt}, {m}, {mt}", "valu"),
        (f"v_sub_f32 {t0}, {m}, {mt}", "valu"),
        (f"v_exp_f32 {al}, {t0}", "trans"),
        (f"v_mov_b32 {m}, {mt}", "valu"),
    ]


def _sm_finish_one(j, buf):
    """One block's finishSM with per-block temps (for zipping)."""
    s = lambda i: sj_reg(buf, j, i)  # noqa: E731
    p0, p1 = PP[j]
    rs = RSJ[j]
    mt, al, l = MT[j], AL[j], OP_L[j]
    c = lambda i: f"v{CP[j] + i}"  # noqa: E731
    L = [(f"v_mov_b32 {rs}, 0", "valu")]
    for i in range(8):
        L += [
            (f"v_fma_f32 {p0}, {s(2 * i)}, {OP_CL2}, -{mt}", "valu"),
            (f"v_fma_f32 {p1}, {s(2 * i + 1)}, {OP_CL2}, -{mt}", "valu"),
            (f"v_exp_f32 {p0}, {p0}", "trans"),
            (f"v_exp_f32 {p1}, {p1}", "trans"),
            # TRANS->reader 1 state: the zip partner's op covers it
            (f"v_cvt_pk_bf16_f32 {c(i)}, {p0}, {p1}", "valu"),
            (f"v_add_f32 {p0}, {p0}, {p1}", "valu"),
            (f"v_add_f32 {rs}, {rs}, {p0}", "valu"),
        ]
    L += [
        (f"v_mov_b32 {p0}, {rs}", "valu"),
        (f"v_mov_b32 {p1}, {rs}", "valu"),
        ("s_nop 0", "nop"),
        (f"v_permlane32_swap_b32 {p0}, {p1}", "valu"),
        (f"v_add_f32 {rs}, {p0}, {p1}", "valu"),
        (f"v_fma_f32 {l}, {l}, {al}, {rs}", "valu"),
        ("s_nop 0", "nop"),
        (f"v_permlane32_swap_b32 {c(0)}, {c(2)}", "valu"),
        (f"v_permlane32_swap_b32 {c(1)}, {c(3)}", "valu"),
        (f"v_permlane32_swap_b32 {c(4)}, {c(6)}", "valu"),
        (f"v_permlane32_swap_b32 {c(5)}, {c(7)}", "valu"),
    ]
    return L


def sm_zip_start(buf):
    return zip_streams(_sm_start_one(0, buf), _sm_start_one(1, buf))


def sm_zip_finish(buf):
    return zip_streams(_sm_finish_one(0, buf), _sm_finish_one(1, buf))


def rescale_call(j, label):
    """Vote-skipped O rescale call site: if any lane's alpha != 1, call the
    per-statement shared rescale subroutine for block j (the 200-instr
    body would blow L1i if inlined at every nb; the vote fires rarely —
    the running max stops moving after early tiles)."""
    return [
        (f"v_cmp_neq_f32 vcc, 1.0, {AL[j]}", "valu"),
        ("s_nop 0", "nop"),  # VALU-writes-VCC -> branch-on-VCCZ guard
        (f"s_cbranch_vccz TA_G6_RSD{label}_{j}", "salu"),
        (f"s_call_b64 s[40:41], TA_G6_RSFN_{j}", "salu"),
        (f"TA_G6_RSD{label}_{j}:", "label"),
    ]


def rescale_fn(j):
    """The shared rescale body: O[j] *= alpha_j via v_accvgpr round trip
    (gfx950 VALU cannot take AGPR operands directly; measured compile
    error). Emitted once per tile statement, jumped over in fall-through."""
    al = AL[j]
    L = [(f"TA_G6_RSFN_{j}:", "label"),
         ("s_nop 11", "nop")]  # PV MFMA D -> accvgpr_read (rare path)
    base = j * 64
    for r in range(0, 64, 4):
        for u in range(4):
            L.append((f"v_accvgpr_read_b32 v{168 + u}, a{base + r + u}", "valu"))
        for u in range(4):
            L.append((f"v_mul_f32 v{168 + u}, v{168 + u}, {al}", "valu"))
        for u in range(4):
            L.append((f"v_accvgpr_write_b32 a{base + r + u}, v{168 + u}", "valu"))
    L.append(("s_nop 1", "nop"))  # accvgpr_write -> MFMA operand guard
    L.append(("s_setpc_b64 s[40:41]", "salu"))
    return L


def rescale_tail():
    """Both rescale subroutines + the jump over them (statement tail)."""
    L = [("s_branch TA_G6_RSEND", "salu")]
    L += rescale_fn(0)
    L += rescale_fn(1)
    L.append(("TA_G6_RSEND:", "label"))
    return L


def pv_stream(nb):
    """PV for 32-key block nb: 16 tr-reads (pipelined, one fragment ahead)
    + 16 MFMAs accumulating into a[]. A = V fragment (two adjacent
    tr-reads = bf16x8), B = c packs (pswap'd), D/C literal AGPRs."""
    L = []
    frags = [(ks, nd) for ks in range(2) for nd in range(4)]

    def emit_reads(fi):
        ks, nd = frags[fi]
        slot = VRSLOT[fi % 3]
        off = nb * 8192 + ks * 4096
        L.append((f"ds_read_b64_tr_b16 {vr(slot, 2)}, {OP_VA(0, nd)} offset:{off}", "ds"))
        L.append((f"ds_read_b64_tr_b16 {vr(slot + 2, 2)}, {OP_VA(1, nd)} offset:{off}", "ds"))

    emit_reads(0)
    emit_reads(1)
    for fi in range(8):
        ks, nd = frags[fi]
        slot = VRSLOT[fi % 3]
        # before MFMA(fi): outstanding = f(fi) + f(fi+1) (4 reads); waiting
        # to <=2 leaves f(fi+1) in flight with f(fi) landed
        if fi < 7:
            L.append(("s_waitcnt lgkmcnt(2)", "wait"))
        else:
            L.append(("s_waitcnt lgkmcnt(0)", "wait"))
        a = vr(slot)
        L.append((f"{MFMA} a[{nd * 16}:{nd * 16 + 15}], {a}, "
                  f"v[{CP[0] + ks * 4}:{CP[0] + ks * 4 + 3}], "
                  f"a[{nd * 16}:{nd * 16 + 15}]", "mfma"))
        L.append((f"{MFMA} a[{64 + nd * 16}:{64 + nd * 16 + 15}], {a}, "
                  f"v[{CP[1] + ks * 4}:{CP[1] + ks * 4 + 3}], "
                  f"a[{64 + nd * 16}:{64 + nd * 16 + 15}]", "mfma"))
        if fi + 2 < 8:
            emit_reads(fi + 2)
    return L


# ---------------------------------------------------------------------------
# v1 tile body: sequential phases per nb (correctness-first)
# ---------------------------------------------------------------------------

def tile_body_v1():
    L = []
    for nb in range(4):
        L += qkt_stream(nb, "A")
        L.append(("s_nop 11", "nop"))   # QKT D -> softmax VALU reader
        L += sm_zip_start("A")
        L += sm_zip_finish("A")
        L += rescale_call(0, f"V1N{nb}")
        L += rescale_call(1, f"V1N{nb}")
        L += pv_stream(nb)
        if nb < 3:
            L.append(("s_nop 11", "nop"))  # PV D -> (rare) rescale reader
    L += rescale_tail()
    return L


# ---------------------------------------------------------------------------
# v2 tile body: the stagger schedule —
#   ramp:   QKT(0) | QKT(1)+SM(0)
#   steady: PV(nb-1)||startSM(nb) -> QKT(nb+1)||finishSM(nb)
#   drain:  finishSM(3) exposed, PV(3)
# ---------------------------------------------------------------------------

def interleave(mfma_stream, fills, max_per_gap=5):
    """Place `fills` into the issue gaps of `mfma_stream`: after each MFMA,
    up to max_per_gap fill items (preserving each stream's own order).
    ds/wait items of the MFMA stream keep their position relative to their
    MFMAs. Left-over fills are appended after the stream."""
    out = []
    fi = 0
    for item in mfma_stream:
        out.append(item)
        if item[1] == "mfma":
            placed = 0
            while fi < len(fills) and placed < max_per_gap:
                out.append(fills[fi])
                fi += 1
                if fills[fi - 1][1] != "nop":
                    placed += 1
    out.extend(fills[fi:])
    return out


def tile_body_v2():
    L = []
    # ramp: QKT(0) -> sjA; QKT(1) -> sjB with SM(0) as fills
    L += qkt_stream(0, "A")
    L.append(("s_nop 11", "nop"))
    sm0 = sm_zip_start("A") + sm_zip_finish("A")
    L += interleave(qkt_stream(1, "B"), sm0)
    L += rescale_call(0, "V2R") + rescale_call(1, "V2R")
    # steady state over nb = 1..3
    for nb in range(1, 4):
        cur = "B" if (nb % 2) else "A"    # sj buffer holding scores(nb)
        nxt = "A" if (nb % 2) else "B"
        # phase X: PV(nb-1) || startSM(nb) both blocks
        startf = sm_zip_start(cur)
        L += interleave(pv_stream(nb - 1), startf)
        L.append(("s_nop 11", "nop"))
        # phase Y: QKT(nb+1) || finishSM(nb); last nb has no QKT(4)
        finishf = sm_zip_finish(cur)
        if nb < 3:
            L += interleave(qkt_stream(nb + 1, nxt), finishf)
        else:
            L += finishf
        L += rescale_call(0, f"V2N{nb}") + rescale_call(1, f"V2N{nb}")
    L.append(("s_nop 1", "nop"))
    L += pv_stream(3)
    L += rescale_tail()
    return L


# ---------------------------------------------------------------------------
# emission
# ---------------------------------------------------------------------------

def render(body_items):
    lines = []
    for text, kind in body_items:
        if kind == "label":
            lines.append(text)      # labels are not tab-indented
        else:
            lines.append(text)
    return "\\n\\t".join(lines)


def clobbers():
    regs = [f'"a{i}"' for i in range(192)]  # O a[0:127] + Q a[128:191]
    regs += [f'"v{i}"' for i in range(144, 256)]
    regs += ['"s40"', '"s41"']  # s_call_b64 return address
    regs += ['"vcc"', '"memory"']
    return ", ".join(regs)


def tile_macro(name, body):
    """One asm statement: counted vmcnt wait (the caller pre-drains with a
    separate vmcnt(0) statement when no next-tile stage is in flight),
    barrier, 4 nb phases, shared rescale subroutines, tail barrier.
    Labels get a per-expansion suffix via %=."""
    body = body.replace("TA_G6_RS", "TA_G6_%=_RS")
    return f'''#define {name}(m0, m1, l0, l1, ka, va, cl2)                        \\
  asm volatile(                                                             \\
      "s_waitcnt vmcnt(16)\\n\\t"                                            \\
      "s_barrier\\n\\t"                                                      \\
      "{body}\\n\\t"                                                         \\
      "s_barrier"                                                           \\
      : "+v"(m0), "+v"(m1), "+v"(l0), "+v"(l1)                              \\
      : "v"(ka[0]), "v"(ka[1]), "v"(ka[2]), "v"(ka[3]), "v"(ka[4]),         \\
        "v"(ka[5]), "v"(ka[6]), "v"(ka[7]),                                 \\
        "v"(va[0]), "v"(va[1]), "v"(va[2]), "v"(va[3]), "v"(va[4]),         \\
        "v"(va[5]), "v"(va[6]), "v"(va[7]),                                 \\
        "v"(cl2)                                                            \\
      : {clobbers()})
'''


def load_q_macros():
    """Prologue: write one Q V8 slice (4 dwords) into its literal AGPRs.
    The s_nop 1 covers the v_accvgpr_write -> MFMA-operand hazard (the
    first QKT is a barrier away, but the guard is free here)."""
    out = []
    for j in range(2):
        for s in range(8):
            b = QBASE + j * 32 + s * 4
            writes = "\\n\\t".join(
                f"v_accvgpr_write_b32 a{b + u}, %{u}" for u in range(4))
            clob = ", ".join(f'"a{b + u}"' for u in range(4))
            out.append(
                f'''#define TA_GEN6_LOAD_Q_J{j}_S{s}(w0, w1, w2, w3)                        \\
  asm volatile("{writes}\\n\\ts_nop 1"                                       \\
               :: "v"(w0), "v"(w1), "v"(w2), "v"(w3) : {clob})
''')
    return "\n".join(out)


def zero_o_macro():
    writes = "\\n\\t".join(
        f"v_accvgpr_write_b32 a{i}, 0" for i in range(128))
    return f'''#define TA_GEN6_ZERO_O()                                          \\
  asm volatile("{writes}" ::: {", ".join(f'"a{i}"' for i in range(128))})
'''


def read_o_macros():
    out = []
    for j in range(2):
        for nd in range(4):
            base = j * 64 + nd * 16
            reads = "\\n\\t".join(
                f"v_accvgpr_read_b32 %{i}, a{base + i}" for i in range(16))
            outs = ", ".join(f'"=v"(f[{i}])' for i in range(16))
            out.append(
                f'''#define TA_GEN6_READ_O_J{j}_ND{nd}(f)                               \\
  asm volatile("s_nop 11\\n\\t{reads}" : {outs})
''')
    return "\n".join(out)


# ---------------------------------------------------------------------------
# probe macros (unit probes in fa_kernels.hip reference these — keep the
# original operand-based forms verified in round 1)
# ---------------------------------------------------------------------------

def probe_qkt_pair_stream():
    L = []
    L.append("ds_read_b128 %2, %4 offset:OFF")
    L.append("ds_read_b128 %3, %5 offset:OFF")
    L.append("s_waitcnt lgkmcnt(1)")
    L.append(f"{MFMA} %0, %2, %12, 0")
    L.append(f"{MFMA} %1, %2, %20, 0")
    for s in range(1, 8):
        kr_cur = 3 if (s % 2) else 2
        kr_nxt = 2 if (s % 2) else 3
        if s + 1 < 8:
            L.append(f"ds_read_b128 %{kr_nxt}, %{4 + s + 1} offset:OFF")
            L.append("s_waitcnt lgkmcnt(1)")
        else:
            L.append("s_waitcnt lgkmcnt(0)")
        L.append(f"{MFMA} %0, %{kr_cur}, %{12 + s}, %0")
        L.append(f"{MFMA} %1, %{kr_cur}, %{20 + s}, %1")
    L.append("s_nop 11")
    return L


def probe_softmax_pack_stream():
    L = []
    T0p, T1p, MTp, RSp, P0p, P1p, M, LR, CL = (
        "%9", "%10", "%11", "%12", "%13", "%14", "%15", "%16", "%33")
    sreg = lambda i: f"%{17 + i}"  # noqa: E731
    L += [f"v_max3_f32 {MTp}, {sreg(0)}, {sreg(1)}, {sreg(2)}",
          f"v_max3_f32 {T0p}, {sreg(3)}, {sreg(4)}, {sreg(5)}",
          f"v_max3_f32 {T1p}, {sreg(6)}, {sreg(7)}, {sreg(8)}",
          f"v_max3_f32 {MTp}, {MTp}, {T0p}, {T1p}",
          f"v_max3_f32 {T0p}, {sreg(9)}, {sreg(10)}, {sreg(11)}",
          f"v_max3_f32 {T1p}, {sreg(12)}, {sreg(13)}, {sreg(14)}",
          f"v_max3_f32 {MTp}, {MTp}, {T0p}, {T1p}",
          f"v_max3_f32 {MTp}, {MTp}, {sreg(15)}, {sreg(15)}"]
    L += [f"v_mov_b32 {T0p}, {MTp}",
          f"v_mov_b32 {T1p}, {MTp}",
          "s_nop 1",
          f"v_permlane32_swap_b32 {T0p}, {T1p}",
          f"v_max_f32 {MTp}, {T0p}, {T1p}",
          f"v_mul_f32 {MTp}, {MTp}, {CL}",
          f"v_max_f32 {MTp}, {M}, {MTp}",
          f"v_sub_f32 {T0p}, {M}, {MTp}",
          f"v_exp_f32 %8, {T0p}",
          f"v_mov_b32 {M}, {MTp}",
          f"v_mov_b32 {RSp}, 0"]
    for i in range(8):
        L += [f"v_fma_f32 {P0p}, {sreg(2 * i)}, {CL}, -{MTp}",
              f"v_fma_f32 {P1p}, {sreg(2 * i + 1)}, {CL}, -{MTp}",
              f"v_exp_f32 {P0p}, {P0p}",
              f"v_exp_f32 {P1p}, {P1p}",
              "s_nop 0",
              f"v_cvt_pk_bf16_f32 %{i}, {P0p}, {P1p}",
              f"v_add_f32 {P0p}, {P0p}, {P1p}",
              f"v_add_f32 {RSp}, {RSp}, {P0p}"]
    L += [f"v_mov_b32 {T0p}, {RSp}",
          f"v_mov_b32 {T1p}, {RSp}",
          "s_nop 1",
          f"v_permlane32_swap_b32 {T0p}, {T1p}",
          f"v_add_f32 {RSp}, {T0p}, {T1p}",
          f"v_fma_f32 {LR}, {LR}, %8, {RSp}"]
    L += ["s_nop 1",
          "v_permlane32_swap_b32 %0, %2",
          "v_permlane32_swap_b32 %1, %3",
          "v_permlane32_swap_b32 %4, %6",
          "v_permlane32_swap_b32 %5, %7"]
    return L


def emit_probe_macros():
    qkt = "\\n\\t".join(probe_qkt_pair_stream())
    bodies = []
    for nb in range(4):
        body = qkt.replace("offset:OFF", f"offset:{nb * 8192}")
        bodies.append(f'''#define TA_GEN6_QKT_PAIR_NB{nb}(sj0, sj1, kr0, kr1, ka, qa, qb)  \\
  asm volatile(                                                             \\
      "{body}"                                                              \\
      : "=&v"(sj0), "=&v"(sj1), "+&v"(kr0), "+&v"(kr1)                          \\
      : "v"(ka[0]), "v"(ka[1]), "v"(ka[2]), "v"(ka[3]), "v"(ka[4]),         \\
        "v"(ka[5]), "v"(ka[6]), "v"(ka[7]),                                 \\
        "a"(qa[0]), "a"(qa[1]), "a"(qa[2]), "a"(qa[3]), "a"(qa[4]),         \\
        "a"(qa[5]), "a"(qa[6]), "a"(qa[7]),                                 \\
        "a"(qb[0]), "a"(qb[1]), "a"(qb[2]), "a"(qb[3]), "a"(qb[4]),         \\
        "a"(qb[5]), "a"(qb[6]), "a"(qb[7]))
''')
    sm = "\\n\\t".join(probe_softmax_pack_stream())
    bodies.append(f'''#define TA_GEN6_SOFTMAX_PACK(c, alpha, m_run, l_run, sv, cl2)             \\
  asm volatile(                                                             \\
      "{sm}"                                                                \\
      : "=&v"(c[0]), "=&v"(c[1]), "=&v"(c[2]), "=&v"(c[3]), "=&v"(c[4]),    \\
        "=&v"(c[5]), "=&v"(c[6]), "=&v"(c[7]), "=&v"(alpha),                \\
        "=&v"(ta_gen6_t0), "=&v"(ta_gen6_t1), "=&v"(ta_gen6_mt),            \\
        "=&v"(ta_gen6_rs), "=&v"(ta_gen6_p0), "=&v"(ta_gen6_p1),            \\
        "+v"(m_run), "+v"(l_run)                                            \\
      : "v"(sv[0]), "v"(sv[1]), "v"(sv[2]), "v"(sv[3]), "v"(sv[4]),         \\
        "v"(sv[5]), "v"(sv[6]), "v"(sv[7]), "v"(sv[8]), "v"(sv[9]),         \\
        "v"(sv[10]), "v"(sv[11]), "v"(sv[12]), "v"(sv[13]), "v"(sv[14]),    \\
        "v"(sv[15]), "v"(cl2))
''')
    return "\n".join(bodies)


# ---------------------------------------------------------------------------
# gen7: the occupancy-2 variant. 8 waves (2/SIMD), ONE 32-row q-block per
# wave — the co-resident partner wave hides dependent-VALU latency and
# MFMA-RAW stalls that the 1-wave/SIMD schedule pays in wall time (measured:
# v2's wall = 8.0k active + 6.3k stall cycles per tile; total per-SIMD issue
# work is the same at occ 2, so the ceiling is the ~8k issue-bound floor).
#
# Register budget at __launch_bounds__(512, 2): 256 per wave, and the
# compiler reserves arch VGPRs >= v128 (measured warning), so every fixed
# arch register sits in v[96:127]: scores v[96:111], c packs v[112:119],
# kr v[120:127] in QKT phases = vr (two 4-reg fragment slots) in PV phases
# (disjoint lifetimes). Temps are operand-allocated scratch ("=&v") from
# the compiler's v0..v95 pool (scalar f32 operands, so %N names work).
# O a[0:63], Q a[64:95]. The LDS double buffer folds into the ds-offset
# immediates (+32768 for buffer 1: max base+imm < 64 KiB fits the 16-bit
# field), so ONE ka/va address set serves both buffers.
# ---------------------------------------------------------------------------

G7_SJ = 96
G7_CP = 112
G7_KR = {0: 120, 1: 124}
G7_VRSLOT = [120, 124, 96]  # third slot borrows score regs (dead in PV)
# operands: outputs first. %0 m, %1 l, %2..%8 temps; inputs %9..%16 ka,
# %17..%24 va, %25 cl2
G7_M, G7_L = "%0", "%1"
G7_T0, G7_T1, G7_MT, G7_RS, G7_P0, G7_P1, G7_AL = (
    "%2", "%3", "%4", "%5", "%6", "%7", "%8")
G7_KA = lambda s: f"%{9 + s}"              # noqa: E731
G7_VA = lambda c, nd: f"%{17 + c * 4 + nd}"   # noqa: E731
G7_CL2 = "%25"


def g7_sj(i):
    return f"v{G7_SJ + i}"


def g7_qkt_stream(nb, bufoff):
    """Single-chain QK^T: 8 ds_read_b128 (TWO ahead: LDS dependent-read
    latency ~50 cyc > one 32-cyc MFMA; the third buffer borrows the c-pack
    registers v[112:115], dead during QKT) + 8 MFMAs, D-as-C."""
    off = bufoff + nb * 8192
    s0 = f"v[{G7_SJ}:{G7_SJ + 15}]"
    bufs = [vr(G7_KR[0]), vr(G7_KR[1]), vr(G7_CP)]
    qreg = lambda s: f"a[{64 + s * 4}:{64 + s * 4 + 3}]"  # noqa: E731
    L = []
    L.append((f"ds_read_b128 {bufs[0]}, {G7_KA(0)} offset:{off}", "ds"))
    L.append((f"ds_read_b128 {bufs[1]}, {G7_KA(1)} offset:{off}", "ds"))
    L.append((f"ds_read_b128 {bufs[2]}, {G7_KA(2)} offset:{off}", "ds"))
    for s in range(8):
        # before MFMA(s): outstanding = reads for s, s+1, s+2 (minus the
        # tail); wait to <= the younger two so read(s) has landed
        L.append((f"s_waitcnt lgkmcnt({min(2, 7 - s)})", "wait"))
        cacc = "0" if s == 0 else s0
        L.append((f"{MFMA} {s0}, {bufs[s % 3]}, {qreg(s)}, {cacc}", "mfma"))
        if s + 3 < 8:
            L.append((f"ds_read_b128 {bufs[(s + 3) % 3]}, {G7_KA(s + 3)} offset:{off}", "ds"))
    return L


def g7_sm_stream():
    """Single-block softmax (start+finish), exp2 domain."""
    s = g7_sj
    t0, t1, mt, rs, p0, p1, al = (G7_T0, G7_T1, G7_MT, G7_RS, G7_P0,
                                  G7_P1, G7_AL)
    m, l, cl = G7_M, G7_L, G7_CL2
    c = lambda i: f"v{G7_CP + i}"  # noqa: E731
    L = [
        (f"v_max3_f32 {mt}, {s(0)}, {s(1)}, {s(2)}", "valu"),
        (f"v_max3_f32 {t0}, {s(3)}, {s(4)}, {s(5)}", "valu"),
        (f"v_max3_f32 {t1}, {s(6)}, {s(7)}, {s(8)}", "valu"),
        (f"v_max3_f32 {mt}, {mt}, {t0}, {t1}", "valu"),
        (f"v_max3_f32 {t0}, {s(9)}, {s(10)}, {s(11)}", "valu"),
        (f"v_max3_f32 {t1}, {s(12)}, {s(13)}, {s(14)}", "valu"),
        (f"v_max3_f32 {mt}, {mt}, {t0}, {t1}", "valu"),
        (f"v_max3_f32 {mt}, {mt}, {s(15)}, {s(15)}", "valu"),
        (f"v_mov_b32 {t0}, {mt}", "valu"),
        (f"v_mov_b32 {t1}, {mt}", "valu"),
        ("s_nop 1", "nop"),
        (f"v_permlane32_swap_b32 {t0}, {t1}", "valu"),
        (f"v_max_f32 {mt}, {t0}, {t1}", "valu"),
        (f"v_mul_f32 {mt}, {mt}, {cl}", "valu"),
        (f"v_max_f32 {mt}, {m}, {mt}", "valu"),
        (f"v_sub_f32 {t0}, {m}, {mt}", "valu"),
        (f"v_exp_f32 {al}, {t0}", "trans"),
        (f"v_mov_b32 {m}, {mt}", "valu"),
        (f"v_mov_b32 {rs}, 0", "valu"),
    ]
    for i in range(8):
        L += [
            (f"v_fma_f32 {p0}, {s(2 * i)}, {cl}, -{mt}", "valu"),
            (f"v_fma_f32 {p1}, {s(2 * i + 1)}, {cl}, -{mt}", "valu"),
            (f"v_exp_f32 {p0}, {p0}", "trans"),
            (f"v_exp_f32 {p1}, {p1}", "trans"),
            ("s_nop 0", "nop"),
            (f"v_cvt_pk_bf16_f32 {c(i)}, {p0}, {p1}", "valu"),
            (f"v_add_f32 {p0}, {p0}, {p1}", "valu"),
            (f"v_add_f32 {rs}, {rs}, {p0}", "valu"),
        ]
    L += [
        (f"v_mov_b32 {t0}, {rs}", "valu"),
        (f"v_mov_b32 {t1}, {rs}", "valu"),
        ("s_nop 1", "nop"),
        (f"v_permlane32_swap_b32 {t0}, {t1}", "valu"),
        (f"v_add_f32 {rs}, {t0}, {t1}", "valu"),
        (f"v_fma_f32 {l}, {l}, {al}, {rs}", "valu"),
        ("s_nop 1", "nop"),
        (f"v_permlane32_swap_b32 {c(0)}, {c(2)}", "valu"),
        (f"v_permlane32_swap_b32 {c(1)}, {c(3)}", "valu"),
        (f"v_permlane32_swap_b32 {c(4)}, {c(6)}", "valu"),
        (f"v_permlane32_swap_b32 {c(5)}, {c(7)}", "valu"),
    ]
    return L


def g7_rescale_call(label):
    return [
        (f"v_cmp_neq_f32 vcc, 1.0, {G7_AL}", "valu"),
        ("s_nop 0", "nop"),
        (f"s_cbranch_vccz TA_G6_RSD{label}", "salu"),
        (f"s_call_b64 s[40:41], TA_G6_RSFN7", "salu"),
        (f"TA_G6_RSD{label}:", "label"),
    ]


def g7_rescale_tail():
    t = [G7_T0, G7_T1, G7_MT, G7_RS]
    L = [("s_branch TA_G6_RSEND", "salu"),
         ("TA_G6_RSFN7:", "label"),
         ("s_nop 11", "nop")]
    for r in range(0, 64, 4):
        for u in range(4):
            L.append((f"v_accvgpr_read_b32 {t[u]}, a{r + u}", "valu"))
        for u in range(4):
            L.append((f"v_mul_f32 {t[u]}, {t[u]}, {G7_AL}", "valu"))
        for u in range(4):
            L.append((f"v_accvgpr_write_b32 a{r + u}, {t[u]}", "valu"))
    L.append(("s_nop 1", "nop"))
    L.append(("s_setpc_b64 s[40:41]", "salu"))
    L.append(("TA_G6_RSEND:", "label"))
    return L


def g7_pv_stream(nb, bufoff):
    """PV: 16 tr-reads (two 4-reg slots, one fragment of lookahead) +
    8 MFMAs into a[0:63]. The slots reuse kr's v[120:127] (QKT-only)."""
    L = []
    frags = [(ks, nd) for ks in range(2) for nd in range(4)]

    def emit_reads(fi):
        ks, nd = frags[fi]
        slot = G7_VRSLOT[fi % 3]
        off = bufoff + nb * 8192 + ks * 4096
        L.append((f"ds_read_b64_tr_b16 {vr(slot, 2)}, {G7_VA(0, nd)} offset:{off}", "ds"))
        L.append((f"ds_read_b64_tr_b16 {vr(slot + 2, 2)}, {G7_VA(1, nd)} offset:{off}", "ds"))

    emit_reads(0)
    emit_reads(1)
    for fi in range(8):
        ks, nd = frags[fi]
        slot = G7_VRSLOT[fi % 3]
        if fi + 2 < 8:
            emit_reads(fi + 2)
            L.append(("s_waitcnt lgkmcnt(4)", "wait"))
        elif fi + 2 == 8:
            L.append(("s_waitcnt lgkmcnt(2)", "wait"))
        else:
            L.append(("s_waitcnt lgkmcnt(0)", "wait"))
        a = vr(slot)
        L.append((f"{MFMA} a[{nd * 16}:{nd * 16 + 15}], {a}, "
                  f"v[{G7_CP + ks * 4}:{G7_CP + ks * 4 + 3}], "
                  f"a[{nd * 16}:{nd * 16 + 15}]", "mfma"))
    return L


def g7_tile_body(bufoff):
    L = []
    for nb in range(4):
        L += g7_qkt_stream(nb, bufoff)
        L.append(("s_nop 11", "nop"))   # QKT D -> softmax VALU reader
        L += g7_sm_stream()
        L += g7_rescale_call(f"N{nb}")
        L += g7_pv_stream(nb, bufoff)
        if nb < 3:
            L.append(("s_nop 11", "nop"))
    L += g7_rescale_tail()
    return L


def g7_clobbers():
    regs = [f'"a{i}"' for i in range(96)]   # O a[0:63] + Q a[64:95]
    regs += [f'"v{i}"' for i in range(96, 128)]
    regs += ['"s40"', '"s41"', '"vcc"', '"memory"']
    return ", ".join(regs)


def g7_tile_macro(name, body):
    body = body.replace("TA_G6_RS", "TA_G6_%=_RS")
    return (
        f"#define {name}(m0, l0, tmp, ka, va, cl2)                          \\\n"
        "  asm volatile(                                                    \\\n"
        '      "s_waitcnt vmcnt(8)\\n\\t"                                    \\\n'
        '      "s_barrier\\n\\t"                                             \\\n'
        f'      "{body}\\n\\t"                                               \\\n'
        '      "s_barrier"                                                  \\\n'
        '      : "+v"(m0), "+v"(l0),                                        \\\n'
        '        "=&v"(tmp[0]), "=&v"(tmp[1]), "=&v"(tmp[2]), "=&v"(tmp[3]),\\\n'
        '        "=&v"(tmp[4]), "=&v"(tmp[5]), "=&v"(tmp[6])                \\\n'
        '      : "v"(ka[0]), "v"(ka[1]), "v"(ka[2]), "v"(ka[3]), "v"(ka[4]),\\\n'
        '        "v"(ka[5]), "v"(ka[6]), "v"(ka[7]),                        \\\n'
        '        "v"(va[0]), "v"(va[1]), "v"(va[2]), "v"(va[3]), "v"(va[4]),\\\n'
        '        "v"(va[5]), "v"(va[6]), "v"(va[7]),                        \\\n'
        '        "v"(cl2)                                                   \\\n'
        f"      : {g7_clobbers()})\n"
    )


def g7_aux_macros():
    parts = []
    writes = "\\n\\t".join(
        f"v_accvgpr_write_b32 a{i}, 0" for i in range(64))
    aclob = ", ".join(f'"a{i}"' for i in range(64))
    parts.append(
        "#define TA_GEN7_ZERO_O()                                         \\\n"
        f'  asm volatile("{writes}" ::: {aclob})\n')
    for s in range(8):
        b = 64 + s * 4
        w = "\\n\\t".join(
            f"v_accvgpr_write_b32 a{b + u}, %{u}" for u in range(4))
        clob = ", ".join(f'"a{b + u}"' for u in range(4))
        parts.append(
            f"#define TA_GEN7_LOAD_Q_S{s}(w0, w1, w2, w3)                  \\\n"
            f'  asm volatile("{w}\\n\\ts_nop 1"                             \\\n'
            f'               :: "v"(w0), "v"(w1), "v"(w2), "v"(w3) : {clob})\n')
    for nd in range(4):
        base = nd * 16
        reads = "\\n\\t".join(
            f"v_accvgpr_read_b32 %{i}, a{base + i}" for i in range(16))
        outs = ", ".join(f'"=v"(f[{i}])' for i in range(16))
        parts.append(
            f"#define TA_GEN7_READ_O_ND{nd}(f)                             \\\n"
            f'  asm volatile("s_nop 11\\n\\t{reads}" : {outs})\n')
    return "\n".join(parts)


def tile_body_ablate(which):
    """Timing-ablation tile bodies (guide §5 common-mistake 8: ablate
    before optimizing). Outputs are garbage — tools-only, env-gated.
    which: "qkt" = staging skeleton + QKT streams only;
           "qktsm" = + zipped softmax; "qktpv" = QKT + PV with constant
           P packs (softmax skipped; c = 1.0 pairs so MFMA data is sane
           for DVFS comparability)."""
    L = []
    for nb in range(4):
        L += qkt_stream(nb, "A")
        L.append(("s_nop 11", "nop"))
        if which == "qktsm":
            L += sm_zip_start("A")
            L += sm_zip_finish("A")
        elif which == "qktpv":
            if nb == 0:
                for j in range(2):
                    for i in range(8):
                        L.append((f"v_mov_b32 v{CP[j] + i}, 0x3f803f80",
                                  "valu"))
            L += pv_stream(nb)
    return L


def stats(items):
    from collections import Counter
    c = Counter(k for _, k in items)
    return dict(c)


def emit():
    parts = ["// GENERATED by tools/gen/gen_prefill6.py — do not edit by hand.",
             "#pragma once", ""]
    parts.append(emit_probe_macros())
    parts.append(zero_o_macro())
    parts.append(load_q_macros())
    parts.append(read_o_macros())
    v1 = tile_body_v1()
    v2 = tile_body_v2()
    parts.append(f"// v1 (sequential phases): {stats(v1)}")
    parts.append(tile_macro("TA_GEN6_TILE_V1", render(v1)))
    parts.append(f"// v2 (staggered schedule): {stats(v2)}")
    parts.append(tile_macro("TA_GEN6_TILE_V2", render(v2)))
    for ver, which in ((3, "qkt"), (4, "qktsm"), (5, "qktpv")):
        body = tile_body_ablate(which)
        parts.append(f"// ablation {which}: {stats(body)}")
        parts.append(tile_macro(f"TA_GEN6_TILE_V{ver}", render(body)))
    parts.append(g7_aux_macros())
    for buf in (0, 1):
        g7 = g7_tile_body(buf * 32768)
        parts.append(f"// gen7 buf{buf} (occ 2, one q-block/wave): {stats(g7)}")
        body = render(g7)
        parts.append(g7_tile_macro(f"TA_GEN7_TILE_B{buf}", body))
        # fp16 variant: same stream, f16 MFMA + RTZ f16 pack (the only
        # dtype-dependent instructions)
        fbody = body.replace("v_mfma_f32_32x32x16_bf16",
                             "v_mfma_f32_32x32x16_f16") \
                    .replace("v_cvt_pk_bf16_f32", "v_cvt_pkrtz_f16_f32")
        parts.append(g7_tile_macro(f"TA_GEN7_TILE_B{buf}_F16", fbody))
    # no-sync ablation: same body, no vmcnt/barriers (garbage output) —
    # isolates the per-tile barrier + DMA-wait share of the wall time
    g8 = g7_tile_body(0)
    parts.append(g7_tile_macro("TA_GEN7_TILE_NOSYNC", render(g8))
                 .replace('"s_waitcnt vmcnt(8)\\n\\t"', '""')
                 .replace('"s_barrier\\n\\t"', '""')
                 .replace('"s_barrier"', '"s_nop 0"'))
    src = "\n".join(parts)
    with open(OUT, "w") as f:
        f.write(src)
    print("wrote", OUT, len(src), "bytes")
    print("v1:", stats(v1), " v2:", stats(v2))
    # the ninja hip rule has no header depfile and hipify caches on content:
    # force the next build to recompile the kernel TU
    hipdir = os.path.dirname(os.path.abspath(OUT))
    for stale in ("fa_kernels_hip.cuda.o", "fa_kernels_hip.hip"):
        f = os.path.join(hipdir, stale)
        if os.path.exists(f):
            os.remove(f)
            print("removed stale", stale)


if __name__ == "__main__":
    emit()
