#!/usr/bin/env python3
"""Memory-lean BERT training: ZeRO v1 gradient/optimizer-state sharding,
optionally stacked with bf16 wire compression and a checkpoint
save/resume (launch: torchrun --nproc-per-node 8
examples/train_bert_zero.py).  At world 1 ZeRO degenerates to plain DP.

Swap "zero.level": "v1" for "offload.level": "v0" to keep optimizer
state in pinned host memory instead (GPU-only; per-bucket D2H overlaps
the backward)."""
import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

import torch
import easyparallellibrary_amd as epl

TINY = os.environ.get("EPL_EXAMPLE_TINY", "0") == "1"
CFG = (dict(layers=2, hidden=128, heads=2, ffn=256) if TINY
       else "bert-large")
BATCH, SEQ = (4, 64) if TINY else (32, 512)
from easyparallellibrary_amd.models import bert
from easyparallellibrary_amd.ops.distributed_losses import ParallelCrossEntropy

epl.init(epl.Config({
    "zero.level": "v1",                    # shard grads + optimizer state
    "communication.compression": "bf16",   # cast the allreduce wire
}))
model = bert.build_bert(CFG)
engine = epl.Engine(model, loss_fn=ParallelCrossEntropy(),
                    optimizer="adamw", lr=1e-4,
                    dtype=torch.bfloat16 if torch.cuda.is_available()
                    else torch.float32)
for step in range(6):
    ids, tgt = bert.synthetic_mlm_batch(BATCH, SEQ, device=engine.device,
                                        seed=step)
    loss = engine.train_step(ids, tgt)
    if engine.rank == 0:
        print("step", step, "loss", float(loss))

# sharded checkpoint: every rank writes its shard-sized optimizer state,
# 50 MB-bucketed part files; restore reshards if the world size changed
ckpt = os.path.join(tempfile.gettempdir(), "bert_zero_ckpt")
engine.save_checkpoint(ckpt)
engine.load_checkpoint(ckpt)
for step in range(6, 10):
    ids, tgt = bert.synthetic_mlm_batch(BATCH, SEQ, device=engine.device,
                                        seed=step)
    loss = engine.train_step(ids, tgt)
    if engine.rank == 0:
        print("step", step, "loss", float(loss))
