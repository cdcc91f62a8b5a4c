#!/usr/bin/env python3
"""Data-parallel BERT training (launch: python -m
easyparallellibrary_amd.launcher --num_workers 8 examples/train_bert_dp.py
or torchrun --nproc-per-node 8 ...)."""
import os
import sys

# allow running as a plain script from anywhere in the repo
sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

import torch
import easyparallellibrary_amd as epl

# EPL_EXAMPLE_TINY=1 shrinks everything for a CPU smoke run
TINY = os.environ.get("EPL_EXAMPLE_TINY", "0") == "1"
CFG = (dict(layers=2, hidden=128, heads=2, ffn=256) if TINY
       else "bert-large")
BATCH, SEQ = (4, 64) if TINY else (32, 512)
from easyparallellibrary_amd.models import bert
from easyparallellibrary_amd.ops.distributed_losses import ParallelCrossEntropy

epl.init(epl.Config({"zero.level": ""}))
model = bert.build_bert(CFG)
engine = epl.Engine(model, loss_fn=ParallelCrossEntropy(),
                    optimizer="adamw", lr=1e-4,
                    dtype=torch.bfloat16 if torch.cuda.is_available()
                    else torch.float32)
for step in range(10):
    ids, tgt = bert.synthetic_mlm_batch(BATCH, SEQ, device=engine.device,
                                        seed=step)
    loss = engine.train_step(ids, tgt)
    merged = engine.all_reduce_metric(loss)   # collective: every rank
    if engine.rank == 0:
        print("step", step, "loss", float(merged))
