#!/usr/bin/env python3
"""Sequence-parallel (ring attention) causal LM: each rank holds a
sequence shard; K/V blocks rotate the ring; parameters replicate and
gradients DP-average across the group.

Launch: torchrun --nproc-per-node N examples/train_long_context.py
(N=1 runs serially).  EPL_EXAMPLE_TINY=1 shrinks it for a CPU smoke."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

import torch
import torch.nn as nn

import easyparallellibrary_amd as epl
from easyparallellibrary_amd.comm.backend import create_communicator
from easyparallellibrary_amd.env import Env
from easyparallellibrary_amd.ops.ring_attention import RingSelfAttention

TINY = os.environ.get("EPL_EXAMPLE_TINY", "0") == "1"
VOCAB, HIDDEN, HEADS, LAYERS = ((256, 128, 4, 2) if TINY
                                else (32000, 1024, 16, 12))
SEQ, BATCH = (128, 2) if TINY else (32768, 1)   # FULL sequence length


class Block(nn.Module):
    def __init__(self, comm):
        super().__init__()
        self.ln1 = nn.LayerNorm(HIDDEN)
        self.attn = RingSelfAttention(HIDDEN, HEADS, comm=comm,
                                      causal=True)
        self.ln2 = nn.LayerNorm(HIDDEN)
        self.mlp = nn.Sequential(nn.Linear(HIDDEN, 4 * HIDDEN), nn.GELU(),
                                 nn.Linear(4 * HIDDEN, HIDDEN))

    def forward(self, x):
        x = x + self.attn(self.ln1(x))
        return x + self.mlp(self.ln2(x))


class LongLM(nn.Module):
    def __init__(self, comm):
        super().__init__()
        self.emb = nn.Embedding(VOCAB, HIDDEN)
        self.blocks = nn.ModuleList(Block(comm) for _ in range(LAYERS))
        self.head = nn.Linear(HIDDEN, VOCAB, bias=False)

    def forward(self, ids):
        x = self.emb(ids)
        for b in self.blocks:
            x = b(x)
        return self.head(x)


def lm_loss(logits, targets):
    return nn.functional.cross_entropy(logits.reshape(-1, VOCAB), targets)


epl.init()
Env.get().get_or_create_process_group()
world = int(os.environ.get("WORLD_SIZE", "1"))
rank = int(os.environ.get("RANK", "0"))
sp = create_communicator("sp", list(range(world))) if world > 1 else None
torch.manual_seed(0)
with epl.replicate(1):
    model = LongLM(sp)
engine = epl.Engine(model, loss_fn=lm_loss, optimizer="adamw", lr=1e-4,
                    dtype=torch.bfloat16 if torch.cuda.is_available()
                    else torch.float32)
sl = SEQ // world
lo = rank * sl
for step in range(10):
    g = torch.Generator().manual_seed(step)  # same data on every rank
    ids = torch.randint(0, VOCAB, (BATCH, SEQ), generator=g)
    tgt = torch.randint(0, VOCAB, (BATCH, SEQ), generator=g)
    loss = engine.train_step(ids[:, lo:lo + sl].to(engine.device),
                             tgt[:, lo:lo + sl].reshape(-1)
                             .to(engine.device))
    merged = engine.all_reduce_metric(loss)   # collective: every rank
    if rank == 0:
        print("step", step, "loss", float(merged))
