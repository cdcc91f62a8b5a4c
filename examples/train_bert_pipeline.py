#!/usr/bin/env python3
"""2-stage pipeline x DP BERT (needs world_size a multiple of 2)."""
import torch
import easyparallellibrary_amd as epl
from easyparallellibrary_amd.models import bert
from easyparallellibrary_amd.ops.distributed_losses import ParallelCrossEntropy

epl.init(epl.Config({"pipeline.num_micro_batch": 4}))
model = bert.build_bert("bert-large", num_stages=2)
engine = epl.Engine(model, loss_fn=ParallelCrossEntropy(),
                    optimizer="adamw", lr=1e-4,
                    dtype=torch.bfloat16 if torch.cuda.is_available()
                    else torch.float32)
for step in range(10):
    ids, tgt = bert.synthetic_mlm_batch(32, 512, device=engine.device,
                                        seed=step)
    loss = engine.train_step(ids, tgt)
    if loss is not None:
        print("step", step, "loss", float(loss))
