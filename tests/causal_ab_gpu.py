"""Causal attention A/B: 128-row kernel vs 64-row c64 variant
(EPL_ATTN_CAUSAL_V2 — set by the caller; the flag is read once per
process).  GPT-2 shapes.  Compares against SDPA (AOTriton) too."""
import os, sys, time, math
sys.path.insert(0, "/root/repo")
import torch
import torch.nn.functional as F
import easyparallellibrary_amd as epl
epl.init()
from easyparallellibrary_amd.ops.attention import _FlashAttention

print("EPL_ATTN_CAUSAL_V2 =", os.environ.get("EPL_ATTN_CAUSAL_V2", "<unset>"))

def bench(fn, iters=30, warm=10):
    for _ in range(warm): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us

for (b, h, s) in [(16, 25, 1024), (4, 25, 4096)]:
    d = 64
    scale = 1 / math.sqrt(d)
    q = torch.randn(b, h, s, d, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    k = torch.randn_like(q, requires_grad=True)
    v = torch.randn_like(q, requires_grad=True)
    dout = torch.randn_like(q)
    flops_fwd = 4 * b * h * s * s * d * 0.5

    t = bench(lambda: _FlashAttention.apply(q, k, v, True, scale))
    print(f"b{b} h{h} s{s} causal: ours fwd {t:8.1f} us  {flops_fwd/t/1e6:7.1f} TF")
    t = bench(lambda: F.scaled_dot_product_attention(
        q, k, v, is_causal=True, scale=scale))
    print(f"  sdpa fwd {t:8.1f} us  {flops_fwd/t/1e6:7.1f} TF")

    def ours_fb():
        out = _FlashAttention.apply(q, k, v, True, scale)
        out.backward(dout)
    t = bench(ours_fb, iters=20)
    print(f"  ours f+b {t:8.1f} us")
    def sdpa_fb():
        out = F.scaled_dot_product_attention(q, k, v, is_causal=True,
                                             scale=scale)
        out.backward(dout)
    t = bench(sdpa_fb, iters=20)
    print(f"  sdpa f+b {t:8.1f} us")
