#!/usr/bin/env python3
"""Minimal driver for rocprofv3 counter capture on the prefill kernels.

Env: TREE_ATTN_PREFILL6=1|2 (or unset for prefill2). Args: tq tkv causal
iters (defaults 4096 4096 0 3).
"""
import sys

sys.path.insert(0, "/root/repo")
import torch  # noqa: E402

from tree_attention_torch_amd.ops.flash import local_attention  # noqa: E402

tq = int(sys.argv[1]) if len(sys.argv) > 1 else 4096
tkv = int(sys.argv[2]) if len(sys.argv) > 2 else 4096
causal = bool(int(sys.argv[3])) if len(sys.argv) > 3 else False
iters = int(sys.argv[4]) if len(sys.argv) > 4 else 3

torch.manual_seed(0)
q = torch.randn(1, 32, tq, 128, device="cuda").bfloat16()
k = torch.randn(1, 32, tkv, 128, device="cuda").bfloat16()
v = torch.randn(1, 32, tkv, 128, device="cuda").bfloat16()
for _ in range(3):
    local_attention(q, k, v, is_causal=causal, q_offset=tkv - tq)
torch.cuda.synchronize()
import time  # noqa: E402

t0 = time.perf_counter()
for _ in range(iters):
    local_attention(q, k, v, is_causal=causal, q_offset=tkv - tq)
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / iters
pairs = tq * (tkv - tq) + tq * (tq + 1) // 2 if causal else tq * tkv
print("%.3f ms  %.1f TF/s" % (dt * 1e3, 2 * 2 * 32 * pairs * 128 / dt / 1e12))
