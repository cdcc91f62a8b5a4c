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

# EPL_EXAMPLE_TINY=1 shrinks everything for a CPU smoke run
TINY = os.environ.get("EPL_EXAMPLE_TINY", "0") == "1"

epl.init(epl.Config({
    "cluster.colocate_split_and_replicate": True,
    # small-batch MoE steps are host-launch-bound: capture fwd+bwd into
    # a hipGraph and replay (single-rank only; multi-rank runs fall
    # back to eager inside the engine with a logged reason)
    "kernel.hip_graph": True,
}))
world = int(os.environ.get("WORLD_SIZE", "1"))
if TINY:
    model = build_moe_transformer(world=world, layers=2, hidden=64, heads=2,
                                  ffn=128, num_experts=max(2, world),
                                  vocab_size=512, max_pos=64)
else:
    model = build_moe_transformer(world=world, layers=12, hidden=1024,
                                  heads=16, ffn=4096,
                                  num_experts=max(8, world))
engine = epl.Engine(model, loss_fn=ParallelCrossEntropy(),
                    optimizer="adamw", lr=1e-4,
                    dtype=torch.bfloat16 if torch.cuda.is_available()
                    else torch.float32)
for step in range(10):
    ids, tgt = gpt2.synthetic_lm_batch(*((2, 32, 512) if TINY else (8, 1024, 32000)),
                                       device=engine.device, seed=step)
    loss = engine.train_step(ids, tgt)
    if engine.rank == 0:
        print("step", step, "loss", float(loss))
