"""Dense Megatron-TP transformer with pipeline stages.

``build_tp_pipeline`` mirrors ``moe_transformer.build_moe_pipeline``:
stage ``s`` is a ``replicate(tp)`` scope over ``tp`` ranks; its
attention/MLP shards live in a per-stage ``split(tp)`` scope that
colocates with the stage's ranks, so every TP all-reduce stays inside
the stage and activations cross stage boundaries as full tensors
(position-wise p2p chains, parallel/pipeline.py).
Needs world = stages * tp * replicas.
"""

import torch.nn as nn

import easyparallellibrary_amd as epl
from easyparallellibrary_amd.ops.tp_mlp import (TensorParallelMLP,
                                                TensorParallelSelfAttention)


class TPBlock(nn.Module):
    """Pre-LN transformer block over deferred-sharding TP modules."""

    def __init__(self, hidden, heads, ffn, causal=True):
        super().__init__()
        self.ln1 = nn.LayerNorm(hidden)
        self.attn = TensorParallelSelfAttention(hidden, heads,
                                                causal=causal)
        self.ln2 = nn.LayerNorm(hidden)
        self.mlp = TensorParallelMLP(hidden, ffn)

    def forward(self, x):
        x = x + self.attn(self.ln1(x))
        return x + self.mlp(self.ln2(x))


class TPStage(nn.Module):
    def __init__(self, blocks, embeddings=None, head=None):
        super().__init__()
        self.embeddings = embeddings
        self.blocks = blocks
        self.head = head

    def forward(self, x):
        if self.embeddings is not None:
            x = self.embeddings(x)
        for b in self.blocks:
            x = b(x)
        if self.head is not None:
            x = self.head(x)
        return x


class TPPipelineModel(nn.Module):
    def __init__(self, stages):
        super().__init__()
        self.stages = nn.ModuleList(stages)

    def forward(self, ids):
        x = ids
        for s in self.stages:
            x = s(x)
        return x


def build_tp_pipeline(stages=2, tp=1, layers=4, hidden=512, heads=8,
                      ffn=2048, vocab_size=32000, max_pos=1024,
                      causal=True):
    """``tp`` may be one int (every stage gets that TP degree) or a
    per-stage list, e.g. ``tp=[1, 2]`` for a width-1 first stage feeding
    a dense-TP-2 second stage (mixed-width pipeline).  Dense-TP stages
    keep boundary activations replicated across their positions, so
    wide stages are declared ``replicated_io`` and the pipeline runtime
    fans activations/grads out/in at 1<->k boundaries
    (parallel/pipeline.py _init_mixed).  Needs world =
    sum(stage widths) * replicas."""
    assert layers % stages == 0, "layers must divide evenly into stages"
    per = layers // stages
    tps = (list(tp) if isinstance(tp, (list, tuple))
           else [tp] * stages)
    assert len(tps) == stages, "per-stage tp list must have one entry " \
        "per stage"
    stage_mods = []
    for s in range(stages):
        # split scopes cannot open inside an explicit replicate scope
        # (reference nesting rule): build the stage shell first, attach
        # the TP blocks from a sibling split scope
        with epl.replicate(tps[s], name="stage_{}".format(s),
                           replicated_io=tps[s] > 1):
            emb = nn.Embedding(vocab_size, hidden) if s == 0 else None
            blocks = nn.ModuleList()
            head = (nn.Linear(hidden, vocab_size, bias=False)
                    if s == stages - 1 else None)
            stage_mods.append(TPStage(blocks, emb, head))
        with epl.split(device_count=tps[s], name="tp_{}".format(s)):
            for _ in range(per):
                blocks.append(TPBlock(hidden, heads, ffn, causal=causal))
    return TPPipelineModel(stage_mods)
