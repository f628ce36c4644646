"""Epilogue latency: HIP combine_packed vs the eager chain, decode shapes."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from tree_attention_torch_amd.ops import flash
from tree_attention_torch_amd.parallel.combine import combine_partials

ext = flash._load_extension()
dev = "cuda"

def t_us(f, steps=200, warm=20):
    for _ in range(warm):
        f()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(steps):
        f()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / steps * 1e6

for s, b, h, tq, d in [(8, 1, 32, 1, 128), (8, 1, 32, 16, 128), (4, 1, 32, 1, 128)]:
    outs = torch.randn(s, b, h, tq, d, device=dev)
    lses = torch.randn(s, b, h, tq, device=dev) * 4
    packed = torch.cat([outs, lses.unsqueeze(-1)], dim=-1).contiguous().view(-1)
    a = t_us(lambda: ext.combine_packed(packed, s, b, h, tq, d))
    e = t_us(lambda: combine_partials(outs, lses))
    print(f"S={s} B={b} H={h} Tq={tq} D={d}: kernel {a:.1f} us  eager {e:.1f} us")
