"""Same-box A/B: chunked vs unchunked single-GPU causal prefill."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from tree_attention_torch_amd.parallel.tree import tree_attention
from tree_attention_torch_amd.data import make_data

dev = torch.device("cuda:0")

def bench(h, t, chunk, steps=6, warm=2):
    q, k, v = make_data((1, h, t, 128), 0, dev, q_len=t, dtype="bf16")
    f = lambda: tree_attention(q, k, v, is_causal=True, q_chunk=chunk)
    for _ in range(warm):
        f()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(steps):
        f()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / steps * 1e3

import sys
chunks = [int(x) for x in sys.argv[1:]] or None
for h, t in [(32, 32768), (8, 16384), (32, 65536)]:
    cks = chunks or [t, max(4096, (512 * 256) // h)]
    for rep in range(2):
        line = f"H={h} T={t} rep{rep}:"
        for ck in cks:
            ms = bench(h, t, min(ck, t))
            line += f"  chunk{min(ck, t)}={ms:.2f}ms"
        print(line, flush=True)
