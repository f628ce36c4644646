"""Serving-step latency: eager tree_attention vs DecodeSession (cache kernel)
vs hipGraph-replayed session step, at short decode lengths where launch
overhead dominates."""
import sys, time, torch
sys.path.insert(0, "/root/repo")
import tree_attention_torch_amd as ta
from tree_attention_torch_amd.ops.flash import local_attention

torch.manual_seed(0)
b, h, d = 1, 32, 128
for t in (4096, 32768):
    q = torch.randn(b, h, 1, d, device="cuda").bfloat16()
    k = torch.randn(b, h, t, d, device="cuda").bfloat16()
    v = torch.randn(b, h, t, d, device="cuda").bfloat16()
    sess = ta.DecodeSession(b, h, d, max_tokens=t, device="cuda")
    sess.prefill(k, v)
    sess.sync_len()
    def timeit(fn, n=200):
        for _ in range(20): fn()
        torch.cuda.synchronize(); t0 = time.perf_counter()
        for _ in range(n): fn()
        torch.cuda.synchronize(); return (time.perf_counter() - t0) / n * 1e6
    us_eager = timeit(lambda: local_attention(q, k, v))
    us_sess = timeit(lambda: sess.attend(q))
    replay, out = sess.graphed_attend(q)
    us_graph = timeit(replay)
    print(f"seq {t}: eager {us_eager:7.1f} us  session {us_sess:7.1f} us  graphed {us_graph:7.1f} us")
