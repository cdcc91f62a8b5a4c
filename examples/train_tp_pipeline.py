#!/usr/bin/env python3
"""Dense Megatron-TP pipeline stages (PP x TP), including mixed stage
widths.

With world=4: 2 stages x TP2 (every stage sharded 2-way, activations
cross stage boundaries as full tensors over per-position p2p chains).
With world=3: mixed widths [1, 2] — a plain width-1 embedding stage
feeding a dense-TP-2 stage; the pipeline runtime fans activations out
1->2 and takes the already-all-reduced input-grad back from position 0
(parallel/pipeline.py _init_mixed).
"""
import os
import sys

# allow running as a plain script from anywhere in the repo
sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

import torch
import torch.nn as nn
import easyparallellibrary_amd as epl
from easyparallellibrary_amd.models.tp_transformer import build_tp_pipeline

# EPL_EXAMPLE_TINY=1 shrinks everything for a CPU smoke run
TINY = os.environ.get("EPL_EXAMPLE_TINY", "0") == "1"

epl.init(epl.Config({"cluster.colocate_split_and_replicate": True,
                     "pipeline.num_micro_batch": 4}))
world = int(os.environ.get("WORLD_SIZE", "1"))
if world == 3:
    tp = [1, 2]          # mixed-width: narrow stage 0, TP-2 stage 1
elif world % 2 == 0 and world > 1:
    tp = world // 2      # 2 stages, each TP world/2
else:
    tp = 1               # degrade to plain 2-stage PP (or serial)

dims = (dict(layers=2, hidden=64, heads=4, ffn=128, vocab_size=512,
             max_pos=64) if TINY
        else dict(layers=24, hidden=1024, heads=16, ffn=4096,
                  vocab_size=32000, max_pos=1024))
# a 2-stage pipeline needs >= 2 ranks; at world 1 run the same model
# as one serial stage so the script stays runnable everywhere
model = build_tp_pipeline(stages=2 if world > 1 else 1, tp=tp, **dims)
V = dims["vocab_size"]


def lm_loss(logits, targets):
    return nn.functional.cross_entropy(
        logits.float().reshape(-1, V), targets.reshape(-1))


engine = epl.Engine(model, loss_fn=lm_loss, optimizer="adamw", lr=1e-4,
                    dtype=torch.bfloat16 if torch.cuda.is_available()
                    else torch.float32)
B, S = (4, 32) if TINY else (16, 512)
for step in range(10):
    g = torch.Generator().manual_seed(1000 + step)
    ids = torch.randint(0, V, (B, S), generator=g).to(engine.device)
    tgt = torch.randint(0, V, (B * S,), generator=g).to(engine.device)
    loss = engine.train_step(ids, tgt)
    if loss is not None:
        print("step", step, "loss", float(loss))
