"""colsum vs torch.sum micro-bench (script, GPU box)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import easyparallellibrary_amd as epl
from easyparallellibrary_amd.ops.bias_linear import fused_colsum

epl.init()


def bench(fn, iters=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


for cols in (1024, 3072):
    dy = torch.randn(65536, cols, device="cuda", dtype=torch.bfloat16)
    tk = bench(lambda: fused_colsum(dy))
    tt = bench(lambda: dy.sum(dim=0))
    gb = 65536 * cols * 2 / 1e9
    print(f"[65536,{cols}] ours {tk:7.1f} us ({gb/tk*1e6:5.2f} TB/s)  "
          f"torch {tt:7.1f} us ({gb/tt*1e6:5.2f} TB/s)")
