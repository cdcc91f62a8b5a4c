"""Op-level (torch.profiler) attribution of the MoE step: which ATen
ops own the dispatch-path time (the kernel trace shows a 9 ms/step
elementwise + sort/index/fill cluster)."""
import sys
sys.path.insert(0, "/root/repo")
import torch
from torch.profiler import ProfilerActivity, profile

import easyparallellibrary_amd as epl

epl.init(epl.Config({"cluster.colocate_split_and_replicate": True}))
from easyparallellibrary_amd.models.moe_transformer import (
    build_moe_transformer)
from easyparallellibrary_amd.ops.distributed_losses import (
    ParallelCrossEntropy)
from easyparallellibrary_amd.models import gpt2

epl_cfg = None
model = build_moe_transformer(world=1, layers=12, hidden=1024, heads=16,
                              ffn=4096, num_experts=8, vocab_size=32000,
                              max_pos=1024)
engine = epl.Engine(model, loss_fn=ParallelCrossEntropy(),
                    optimizer="adamw", lr=1e-4, dtype=torch.bfloat16)
ids, tgt = gpt2.synthetic_lm_batch(8, 1024, 32000, device=engine.device,
                                   seed=1)
for _ in range(3):
    engine.train_step(ids, tgt)
torch.cuda.synchronize()
with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA]) as p:
    for _ in range(2):
        engine.train_step(ids, tgt)
    torch.cuda.synchronize()
print(p.key_averages().table(sort_by="cuda_time_total", row_limit=30))
