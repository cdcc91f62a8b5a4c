"""MoE transformer (reference tutorial: docs/en/tutorials/moe.md —
default strategy replicate(world) for attention/gating, expert weights in
epl.split(world) with all-to-all dispatch/combine)."""

import torch
import torch.nn as nn

import easyparallellibrary_amd as epl
from easyparallellibrary_amd.models.transformer import (Embeddings, LMHead,
                                                        SelfAttention,
                                                        init_weights)
from easyparallellibrary_amd.ops.layer_norm import FusedLayerNorm
from easyparallellibrary_amd.ops.moe import ExpertParallelMLP


class MoEBlock(nn.Module):
    def __init__(self, hidden, heads, ffn, num_experts, split_degree=1,
                 top_k=2, make_moe=True):
        super().__init__()
        self.ln1 = FusedLayerNorm(hidden)
        self.attn = SelfAttention(hidden, heads, causal=True)
        self.ln2 = FusedLayerNorm(hidden)
        if make_moe:
            with epl.split(device_count=split_degree, name="experts"):
                self.moe = ExpertParallelMLP(hidden, ffn, num_experts,
                                             top_k=top_k)
        else:
            self.moe = None  # attached later (pipeline builder)

    def forward(self, x):
        a = self.attn(self.ln1(x))
        y, s = self.ln2.forward_with_sum(a, x)
        return s + self.moe(y)


class MoETransformer(nn.Module):
    def __init__(self, layers=4, hidden=512, heads=8, ffn=2048,
                 num_experts=8, vocab_size=32000, max_pos=1024, top_k=2,
                 split_degree=1):
        super().__init__()
        self.embeddings = Embeddings(vocab_size, hidden, max_pos,
                                     use_ln=False)
        self.blocks = nn.ModuleList(
            MoEBlock(hidden, heads, ffn, num_experts, split_degree, top_k)
            for _ in range(layers))
        self.head = LMHead(hidden, vocab_size)

    def forward(self, ids):
        x = self.embeddings(ids)
        for b in self.blocks:
            x = b(x)
        return self.head(x)


class MoEStage(nn.Module):
    """One pipeline stage of a MoE transformer: optional embeddings,
    a run of blocks, optional LM head."""

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


class MoEPipelineModel(nn.Module):
    def __init__(self, stages):
        super().__init__()
        self.stages = nn.ModuleList(stages)

    def forward(self, ids):
        x = ids
        for s in self.stages:
            x = s(x)
        return x


def build_moe_pipeline(stages=2, ep=1, layers=4, hidden=512, heads=8,
                       ffn=2048, num_experts=8, vocab_size=32000,
                       max_pos=1024, top_k=2):
    """PP x (DP+EP) hybrid: stage ``s`` is a ``replicate(ep)`` scope over
    ``ep`` ranks (positions act as data-parallel streams); its expert
    weights live in a per-stage ``split(ep)`` scope that colocates with
    the stage's ranks (all-to-all stays inside the stage).  Activations
    move position-wise between stages (parallel/pipeline.py per-position
    chains).  Needs world = stages * ep * replicas."""
    assert layers % stages == 0, "layers must divide evenly into stages"
    per = layers // stages
    stage_mods = []
    for s in range(stages):
        # split scopes cannot open inside an explicit replicate scope
        # (reference nesting rule), so blocks defer their MoE and the
        # experts attach from a sibling split scope.
        with epl.replicate(ep, name="stage_{}".format(s)):
            emb = (Embeddings(vocab_size, hidden, max_pos, use_ln=False)
                   if s == 0 else None)
            blocks = nn.ModuleList(
                MoEBlock(hidden, heads, ffn, num_experts, top_k=top_k,
                         make_moe=False)
                for _ in range(per))
            head = LMHead(hidden, vocab_size) if s == stages - 1 else None
            stage_mods.append(MoEStage(blocks, emb, head))
        with epl.split(device_count=ep, name="experts_{}".format(s)):
            for b in blocks:
                b.moe = ExpertParallelMLP(hidden, ffn, num_experts,
                                          top_k=top_k)
    return init_weights(MoEPipelineModel(stage_mods))


def build_moe_transformer(world=None, **kwargs):
    """default strategy = replicate(world) per the reference tutorial; the
    expert weights inside each MoEBlock live in a split scope."""
    if world is None:
        world = epl.Env.get().world_size or 1
    epl.set_default_strategy(epl.replicate(world))
    model = MoETransformer(split_degree=world, **kwargs)
    return init_weights(model)
