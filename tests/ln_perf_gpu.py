import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import easyparallellibrary_amd as epl
epl.init()
from easyparallellibrary_amd import _C

rows, cols = 65536, 1024
x = torch.randn(rows, cols, device="cuda", dtype=torch.bfloat16)
g = torch.randn(cols, device="cuda", dtype=torch.bfloat16)
b = torch.randn(cols, device="cuda", dtype=torch.bfloat16)
dy = torch.randn_like(x)
mean = torch.empty(rows, dtype=torch.float32, device="cuda")
rstd = torch.empty(rows, dtype=torch.float32, device="cuda")
out = torch.empty_like(x)
_C.layer_norm_fwd(out, x, None, None, g, b, mean, rstd, 1e-5)
dx = torch.empty_like(x)
dgamma = torch.zeros(cols, dtype=torch.float32, device="cuda")
dbeta = torch.zeros(cols, dtype=torch.float32, device="cuda")

def run():
    _C.layer_norm_bwd(dx, dgamma, dbeta, dy, x, g, mean, rstd)
for _ in range(10): run()
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(50): run()
torch.cuda.synchronize()
us = (time.perf_counter() - t0)/50*1e6
gb = (rows*cols*2*3 + cols*2) / 1e9
print(f"ln_bwd b128-shape: {us:.1f} us  {gb/us*1e6/1e3:.2f} TB/s")
