#!/usr/bin/env python3
"""Train + EVALUATE under pipeline parallelism: train_step drives the
1F1B schedule, eval_step drives the forward-only pipelined chain
(PipelineRuntime.run_eval — every rank participates; the LAST stage
returns the merged outputs).  Works at any world size (degrades to a
plain engine at world 1); the eval batch size may differ from training
(the eval p2p handshake exchanges shapes per call)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

import torch
import easyparallellibrary_amd as epl

TINY = os.environ.get("EPL_EXAMPLE_TINY", "0") == "1"
CFG = (dict(layers=2, hidden=128, heads=2, ffn=256) if TINY
       else "bert-large")
BATCH, SEQ = (4, 64) if TINY else (32, 512)
EVAL_BATCH = BATCH // 2 or 1

from easyparallellibrary_amd.models import bert
from easyparallellibrary_amd.ops.distributed_losses import ParallelCrossEntropy

epl.init(epl.Config({"pipeline.num_micro_batch": 2 if TINY else 4}))
world = int(os.environ.get("WORLD_SIZE", "1"))
stages = 2 if world % 2 == 0 and world > 1 else 1
model = bert.build_bert(CFG, num_stages=stages)
loss_fn = ParallelCrossEntropy()
engine = epl.Engine(model, loss_fn=loss_fn, optimizer="adamw", lr=1e-4,
                    dtype=torch.bfloat16 if torch.cuda.is_available()
                    else torch.float32)

for step in range(6):
    ids, tgt = bert.synthetic_mlm_batch(BATCH, SEQ, device=engine.device,
                                        seed=step)
    loss = engine.train_step(ids, tgt)

    if step % 3 == 2:
        # pipelined evaluation: COLLECTIVE — every rank calls it; only
        # the last stage gets logits back
        eids, etgt = bert.synthetic_mlm_batch(
            EVAL_BATCH, SEQ, device=engine.device, seed=1000 + step)
        logits = engine.eval_step(eids)
        if logits is not None:
            eval_loss = float(loss_fn(logits, etgt))
            print("step", step, "train",
                  None if loss is None else round(float(loss), 4),
                  "EVAL", round(eval_loss, 4))
        # the reference's eval-barrier signal: every rank syncs on the
        # chief before training resumes (hooks.py:915-933 parity)
        engine.broadcast_signal(0.0, root=0)

engine.close()
print("done")
