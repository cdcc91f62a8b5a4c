#!/usr/bin/env python3
"""Data-parallel BERT training (launch: python -m
easyparallellibrary_amd.launcher --num_workers 8 examples/train_bert_dp.py
or torchrun --nproc-per-node 8 ...)."""
import torch
import easyparallellibrary_amd as epl
from easyparallellibrary_amd.models import bert
from easyparallellibrary_amd.ops.distributed_losses import ParallelCrossEntropy

epl.init(epl.Config({"zero.level": ""}))
model = bert.build_bert("bert-large")
engine = epl.Engine(model, loss_fn=ParallelCrossEntropy(),
                    optimizer="adamw", lr=1e-4,
                    dtype=torch.bfloat16 if torch.cuda.is_available()
                    else torch.float32)
for step in range(10):
    ids, tgt = bert.synthetic_mlm_batch(32, 512, device=engine.device,
                                        seed=step)
    loss = engine.train_step(ids, tgt)
    if engine.rank == 0:
        print("step", step, "loss", float(engine.all_reduce_metric(loss)))
