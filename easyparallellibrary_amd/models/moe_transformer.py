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
                 top_k=2):
        super().__init__()
        self.ln1 = FusedLayerNorm(hidden)
        self.attn = SelfAttention(hidden, heads, causal=True)
        self.ln2 = FusedLayerNorm(hidden)
        with epl.split(device_count=split_degree, name="experts"):
            self.moe = ExpertParallelMLP(hidden, ffn, num_experts,
                                         top_k=top_k)

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


def build_moe_transformer(world=None, **kwargs):
    """default strategy = replicate(world) per the reference tutorial; the
    expert weights inside each MoEBlock live in a split scope."""
    if world is None:
        world = epl.Env.get().world_size or 1
    epl.set_default_strategy(epl.replicate(world))
    model = MoETransformer(split_degree=world, **kwargs)
    return init_weights(model)
