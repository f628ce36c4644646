"""Mixed-workload stability soak: decode + spec-decode + prefill + serving
session, with periodic numerics checks vs the fp32 oracle. Args: seconds."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from tree_attention_torch_amd.ops import flash
from tree_attention_torch_amd.ops.reference import flash_res_lse
from tree_attention_torch_amd.parallel.tree import tree_attention
from tree_attention_torch_amd.session import DecodeSession
from tree_attention_torch_amd.data import make_data

secs = float(sys.argv[1]) if len(sys.argv) > 1 else 120.0
dev = torch.device("cuda:0")
torch.manual_seed(0)

# workloads
q1, k1, v1 = make_data((1, 32, 131072, 128), 0, dev, dtype="bf16")
q2, k2, v2 = make_data((1, 32, 524288, 128), 0, dev, dtype="fp8", kv_heads=4)
qs, ks, vs = make_data((1, 8, 8192, 128), 0, dev, q_len=32, dtype="bf16")
qp, kp, vp = make_data((1, 8, 8192, 128), 0, dev, q_len=8192, dtype="bf16")
sess = DecodeSession(1, 8, 128, max_tokens=1 << 21, device=dev, kv_dtype="bf16")
sess.prefill(ks, vs)
qg = torch.randn(1, 8, 1, 128, device=dev).bfloat16()
replay, out_g = sess.graphed_attend(qg)

# MX serving session (hardware-scale decode over the quantized cache,
# bf16 staging tail, outlier channel alive the whole soak)
mx_sess = DecodeSession(1, 4, 128, max_tokens=1 << 21, device=dev,
                        kv_dtype="mx", block=256)  # 2M-token headroom:
# a 600 s soak appends ~70k tokens (1<<16 overflowed at ~550 s)
kmx = torch.randn(1, 4, 8192, 128, device=dev)
kmx[..., 13] *= 700.0
mx_sess.prefill(kmx, kmx)
qmx = torch.randn(1, 16, 1, 128, device=dev).bfloat16()

t0 = time.time()
it = 0
checks = 0
while time.time() - t0 < secs:
    o1 = tree_attention(q1, k1, v1)
    o2 = tree_attention(q2, k2, v2)
    o3 = tree_attention(qs, ks, vs, is_causal=True)  # spec-decode route
    if it % 20 == 0:
        o4 = tree_attention(qp, kp, vp, is_causal=True)  # full prefill
        assert torch.isfinite(o4).all()
    kn = torch.randn(1, 8, 1, 128, device=dev).bfloat16()
    sess.append(kn, kn)
    sess.sync_len()
    replay()
    assert torch.isfinite(out_g).all()
    if it % 5 == 0:
        kn4 = torch.randn(1, 4, 1, 128, device=dev)
        mx_sess.append(kn4 * (1 + 700.0 * (torch.rand_like(kn4) < 0.01)),
                       kn4)
        omx = mx_sess.attend(qmx)
        assert torch.isfinite(omx).all()
    if it % 50 == 0:
        # numerics spot-check on a fresh small case
        torch.manual_seed(1000 + it)
        qq = torch.randn(1, 4, 1, 128, device=dev).bfloat16()
        kk = torch.randn(1, 4, 3000, 128, device=dev).bfloat16()
        vv = torch.randn(1, 4, 3000, 128, device=dev).bfloat16()
        oo, ll = flash.local_attention(qq, kk, vv)
        ro, rl = flash_res_lse(qq.cpu(), kk.cpu(), vv.cpu())
        torch.testing.assert_close(oo.cpu(), ro, rtol=2.5e-2, atol=2.5e-2)
        checks += 1
    assert torch.isfinite(o1).all() and torch.isfinite(o2).all() \
        and torch.isfinite(o3).all()
    it += 1
    if it % 20000 == 0:
        print(f"  t={time.time()-t0:.0f}s it={it} len={sess.total}", flush=True)
torch.cuda.synchronize()
print(f"SOAK OK: {it} iterations, {checks} oracle spot-checks, "
      f"{time.time()-t0:.0f}s, session len {sess.total}", flush=True)
