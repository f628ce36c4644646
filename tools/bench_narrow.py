"""Narrow-head decode bandwidth: native zero-padded-LDS kernel vs the old
pad-to-128 copy fallback (>3 TB/s on 3 non-{64,128} dims)."""
import sys, time
sys.path.insert(0, "/root/repo")
import torch
import torch.nn.functional as F
from tree_attention_torch_amd.ops import flash

ext = flash._load_extension()
for d in (32, 48, 80, 96, 112):
    torch.manual_seed(0)
    t = 131072
    q = torch.randn(1, 32, 1, d, device="cuda").bfloat16()
    k = torch.randn(1, 32, t, d, device="cuda").bfloat16()
    v = torch.randn(1, 32, t, d, device="cuda").bfloat16()
    for _ in range(5):
        flash.local_attention(q, k, v)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(30):
        flash.local_attention(q, k, v)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 30
    byt = 2 * 32 * t * d * 2
    # old pad-to-128 fallback for comparison
    pad = 128 - d
    def padded():
        qp = F.pad(q, (0, pad)).contiguous()
        kp = F.pad(k, (0, pad)).contiguous()
        vp = F.pad(v, (0, pad)).contiguous()
        o, l = ext.flash_attention(qp, kp, vp, d ** -0.5, False, 0, 0)
        return o[..., :d].contiguous()
    for _ in range(3):
        padded()
    torch.cuda.synchronize(); t1 = time.perf_counter()
    for _ in range(10):
        padded()
    torch.cuda.synchronize()
    dtp = (time.perf_counter() - t1) / 10
    print(f"d={d:3d} 128K decode: native {dt*1e3:.3f} ms ({byt/dt/1e12:.2f} "
          f"TB/s real bytes) | pad-copy fallback {dtp*1e3:.3f} ms "
          f"({dt and dtp/dt:.1f}x)")
