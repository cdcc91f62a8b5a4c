"""Diagnose the causal-vs-SDPA gap: same shapes causal AND non-causal,
so shape effects separate from causal-path effects."""
import os, sys, time, math
sys.path.insert(0, "/root/repo")
import torch
import torch.nn.functional as F
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

for (b, h, s) in [(16, 25, 1024), (4, 25, 4096), (128, 16, 512)]:
    d = 64
    scale = 1 / math.sqrt(d)
    q = torch.randn(b, h, s, d, device="cuda", dtype=torch.bfloat16)
    k = torch.randn_like(q)
    v = torch.randn_like(q)
    for causal in (False, True):
        fl = 4 * b * h * s * s * d * (0.5 if causal else 1.0)
        t = bench(lambda: _FlashAttention.apply(q, k, v, causal, scale))
        t2 = bench(lambda: F.scaled_dot_product_attention(
            q, k, v, is_causal=causal, scale=scale))
        print(f"b{b} h{h} s{s} causal={int(causal)}: "
              f"ours {t:7.1f}us {fl/t/1e6:6.1f}TF | "
              f"sdpa {t2:7.1f}us {fl/t2/1e6:6.1f}TF")
