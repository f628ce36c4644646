"""Speculative-decode shapes (Tq 8..96, long KV): looped split-KV decode vs
the prefill kernel. Emits JSONL for profiles/decode_matrix_1gpu.jsonl."""
import json, os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from tree_attention_torch_amd.ops import flash
from tree_attention_torch_amd.data import make_data

dev = torch.device("cuda:0")
ext = flash._load_extension()
assert ext is not None

def t_ms(f, steps=20, warm=5):
    for _ in range(warm):
        f()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(steps):
        f()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / steps * 1e3

for tkv in (32768, 131072):
    for tq in (8, 16, 17, 32, 64, 96, 128, 256):
        q, k, v = make_data((1, 32, tkv, 128), 0, dev, q_len=tq, dtype="bf16")
        scale = 128 ** -0.5
        off = tkv - tq
        a = t_ms(lambda: flash.local_attention(q, k, v, is_causal=True,
                                               q_offset=off))
        b = t_ms(lambda: ext.flash_attention(q, k, v, scale, True, off, 0))
        print(json.dumps({"case": "spec_decode", "tq": tq, "tkv": tkv,
                          "h": 32, "dispatch_ms": round(a, 4),
                          "prefill_route_ms": round(b, 4)}), flush=True)
