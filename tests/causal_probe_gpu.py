"""Probe the causal-fwd anomaly: batch sweep (grid-size scaling) to test
block-level load-imbalance vs per-tile slowdown."""
import sys, time, math
sys.path.insert(0, "/root/repo")
import torch
import easyparallellibrary_amd as epl
epl.init()
from easyparallellibrary_amd.ops.attention import _FlashAttention

def bench(fn, iters=30, warm=10):
    for _ in range(warm): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6

s, h, d = 1024, 25, 64
scale = 1 / math.sqrt(d)
for b in (2, 4, 16, 64, 128):
    q = torch.randn(b, h, s, d, device="cuda", dtype=torch.bfloat16)
    k = torch.randn_like(q)
    v = torch.randn_like(q)
    tn = bench(lambda: _FlashAttention.apply(q, k, v, False, scale))
    tc = bench(lambda: _FlashAttention.apply(q, k, v, True, scale))
    blocks = (s // 128) * b * h
    print(f"b{b:4d} blocks={blocks:6d}: noncausal {tn:8.1f}us causal "
          f"{tc:8.1f}us  ratio {tc/tn:.3f} (work ratio ~0.56)")
