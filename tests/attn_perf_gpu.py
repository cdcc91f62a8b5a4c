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
    return (time.perf_counter() - t0) / iters * 1e6  # us

for (b, h, s, causal) in [(128, 16, 512, False), (16, 25, 1024, True)]:
    d = 64
    scale = 1/math.sqrt(d)
    q = torch.randn(b, h, s, d, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    k = torch.randn_like(q, requires_grad=True)
    v = torch.randn_like(q, requires_grad=True)
    dout = torch.randn_like(q)
    flops_fwd = 4 * b * h * s * s * d * (0.5 if causal else 1.0)

    # ours fwd
    t = bench(lambda: _FlashAttention.apply(q, k, v, causal, scale))
    print(f"b{b} h{h} s{s} causal={causal}: ours fwd {t:8.1f} us  {flops_fwd/t/1e6:7.1f} TF")
    # sdpa fwd
    t = bench(lambda: F.scaled_dot_product_attention(q, k, v, is_causal=causal, scale=scale))
    print(f"  sdpa fwd {t:8.1f} us  {flops_fwd/t/1e6:7.1f} TF")

    # ours fwd+bwd
    def ours_fb():
        out = _FlashAttention.apply(q, k, v, causal, scale)
        out.backward(dout)
    t = bench(ours_fb, iters=20)
    print(f"  ours f+b {t:8.1f} us  {3.5*flops_fwd/t/1e6:7.1f} TF-equiv")
    def sdpa_fb():
        out = F.scaled_dot_product_attention(q, k, v, is_causal=causal, scale=scale)
        out.backward(dout)
    t = bench(sdpa_fb, iters=20)
    print(f"  sdpa f+b {t:8.1f} us  {3.5*flops_fwd/t/1e6:7.1f} TF-equiv")

# strided-input case: the module path's qkv-unbind views
print("--- strided (qkv unbind views) ---")
for (b, h, s, causal) in [(128, 16, 512, False)]:
    d = 64
    scale = 1/math.sqrt(d)
    qkv = torch.randn(b, s, 3, h, d, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    q, k, v = (t.transpose(1, 2) for t in qkv.unbind(dim=2))
    dout = torch.randn(b, h, s, d, device="cuda", dtype=torch.bfloat16)
    flops_fwd = 4 * b * h * s * s * d

    t = bench(lambda: _FlashAttention.apply(q, k, v, causal, scale))
    print(f"ours fwd strided {t:8.1f} us  {flops_fwd/t/1e6:7.1f} TF")
    t = bench(lambda: F.scaled_dot_product_attention(q, k, v, is_causal=causal, scale=scale))
    print(f"sdpa fwd strided {t:8.1f} us  {flops_fwd/t/1e6:7.1f} TF")
    def ours_fb():
        qkv.grad = None
        out = _FlashAttention.apply(q, k, v, causal, scale)
        out.backward(dout)
    t = bench(ours_fb, iters=20)
    print(f"ours f+b strided {t:8.1f} us")
    def sdpa_fb():
        qkv.grad = None
        out = F.scaled_dot_product_attention(q, k, v, is_causal=causal, scale=scale)
        out.backward(dout)
    t = bench(sdpa_fb, iters=20)
    print(f"sdpa f+b strided {t:8.1f} us")

import os
print("--- bwd split A/B (contiguous) ---")
for split in ("0", "1", "0", "1"):
    os.environ["EPL_ATTN_BWD_SPLIT"] = split
    b, h, s, d = 128, 16, 512, 64
    scale = 1/math.sqrt(d)
    q = torch.randn(b, h, s, d, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    k = torch.randn_like(q, requires_grad=True)
    v = torch.randn_like(q, requires_grad=True)
    dout = torch.randn_like(q)
    def fb():
        out = _FlashAttention.apply(q, k, v, False, scale)
        out.backward(dout)
    t = bench(fb, iters=20)
    print(f"split={split}  f+b {t:8.1f} us")
os.environ.pop("EPL_ATTN_BWD_SPLIT", None)

print("--- module-level qkv paths (BERT shape, fwd+bwd) ---")
from easyparallellibrary_amd.ops.attention import qkv_flash_attention
b, s, h, d = 128, 512, 16, 64
scale = 1/math.sqrt(d)
qkv = torch.randn(b, s, 3, h, d, device="cuda", dtype=torch.bfloat16,
                  requires_grad=True)
dout_bshd = torch.randn(b, s, h, d, device="cuda", dtype=torch.bfloat16)
def fused_fb():
    qkv.grad = None
    out = qkv_flash_attention(qkv, causal=False)
    out.backward(dout_bshd.permute(0, 2, 1, 3))
for _ in range(2):
    t = bench(fused_fb, iters=20)
    print(f"qkv fused native f+b {t:8.1f} us")
def sdpa_split_fb():
    qkv.grad = None
    from easyparallellibrary_amd.models.transformer import _QKVSplit
    q, k, v = _QKVSplit.apply(qkv)
    out = F.scaled_dot_product_attention(q, k, v, is_causal=False, scale=scale)
    out.backward(dout_bshd.transpose(1, 2))
for _ in range(2):
    t = bench(sdpa_split_fb, iters=20)
    print(f"sdpa + qkvsplit  f+b {t:8.1f} us")
