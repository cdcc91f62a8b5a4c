import os, sys, time, math
sys.path.insert(0, "/root/repo")
import torch
import easyparallellibrary_amd as epl
epl.init()
from easyparallellibrary_amd.ops.attention import flash_attention

def fb_time(q, k, v, dout, causal, iters=30, warm=8):
    def fb():
        out = flash_attention(q, k, v, causal=causal)
        out.backward(dout)
        q.grad = k.grad = v.grad = None
    for _ in range(warm): fb()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fb()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6

mode = os.environ.get("EPL_ATTN_BWD_DBUF", "1")
for (b, h, s, causal) in [(128, 16, 512, False), (16, 25, 1024, True),
                          (8, 16, 4096, False)]:
    torch.manual_seed(7)
    q = torch.randn(b, h, s, 64, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    k = torch.randn_like(q, requires_grad=True)
    v = torch.randn_like(q, requires_grad=True)
    dout = torch.randn_like(q)
    t = fb_time(q, k, v, dout, causal)
    print(f"mode={mode} b{b} h{h} s{s} causal={int(causal)}: f+b {t:8.1f} us")
