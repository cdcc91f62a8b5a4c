#!/usr/bin/env python3
"""Expert-parallel MoE transformer (experts split across all ranks)."""
import os
import sys

# allow running as a plain script from anywhere in the repo
sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

import torch
import easyparallellibrary_amd as epl
from easyparallellibrary_amd.models.moe_transformer import build_moe_transformer
from easyparallellibrary_amd.models import gpt2
from easyparallellibrary_amd.ops.distributed_losses import ParallelCrossEntropy

epl.init(epl.Config({"cluster.colocate_split_and_replicate": True}))
world = int(os.environ.get("WORLD_SIZE", "1"))
model = build_moe_transformer(world=world, layers=12, hidden=1024, heads=16,
                              ffn=4096, num_experts=max(8, world))
engine = epl.Engine(model, loss_fn=ParallelCrossEntropy(),
                    optimizer="adamw", lr=1e-4,
                    dtype=torch.bfloat16 if torch.cuda.is_available()
                    else torch.float32)
for step in range(10):
    ids, tgt = gpt2.synthetic_lm_batch(8, 1024, 32000,
                                       device=engine.device, seed=step)
    loss = engine.train_step(ids, tgt)
    if engine.rank == 0:
        print("step", step, "loss", float(loss))
