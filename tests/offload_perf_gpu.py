"""Offload step phase timing harness (run as a script on a GPU box)."""
import os
import time

import torch

print("nproc", os.cpu_count(), "torch threads", torch.get_num_threads())
n = 1 << 28  # 256M elems ~ 1/6 of gpt2-xl arena
w = torch.randn(n, pin_memory=True)
m = torch.zeros(n, pin_memory=True)
v = torch.zeros(n, pin_memory=True)
gc = torch.empty(n, pin_memory=True)
gdev = torch.randn(n, dtype=torch.bfloat16, device="cuda")
pdev = torch.empty(n, dtype=torch.bfloat16, device="cuda")
torch.cuda.synchronize()

for threads in (torch.get_num_threads(), 16, 32, 64, 128):
    torch.set_num_threads(threads)
    t0 = time.time()
    torch._fused_adamw_([w], [gc], [m], [v], [], [torch.tensor(1.0)],
                        lr=1e-3, beta1=0.9, beta2=0.999, weight_decay=0.01,
                        eps=1e-8, amsgrad=False, maximize=False)
    print(f"threads={threads:4d} fused_adamw 256M: {time.time()-t0:.3f}s")

CH = 1 << 24
spans = [(lo, min(lo + CH, n)) for lo in range(0, n, CH)]
s = torch.cuda.Stream()
t0 = time.time()
with torch.cuda.stream(s):
    for lo, hi in spans:
        gc[lo:hi].copy_(gdev[lo:hi].to(torch.float32), non_blocking=True)
s.synchronize()
print(f"D2H convert+copy 256M: {time.time()-t0:.3f}s")
t0 = time.time()
with torch.cuda.stream(s):
    for lo, hi in spans:
        pdev[lo:hi].copy_(w[lo:hi], non_blocking=True)
s.synchronize()
print(f"H2D convert copy 256M: {time.time()-t0:.3f}s")
