"""BERT model family (the flagship benchmark model: BERT-Large MLM on
synthetic data — BASELINE.json configs 2 and 3; the reference's model
zoo is external, see /root/reference/docs/en/tutorials/pipe.md for the
staged-BERT usage this mirrors).

Built with the epl annotation API: ``build_bert(num_stages=S)`` wraps the
layer ranges in ``epl.replicate(name='stage_i')`` scopes so the engine
plans an S-stage pipeline; S=1 is pure DP.  The model is stage-chainable
(stage_i output feeds stage_{i+1}) as the pipeline runtime requires.
"""

import torch
import torch.nn as nn

import easyparallellibrary_amd as epl
from easyparallellibrary_amd.models.transformer import (Block, Embeddings,
                                                        LMHead, init_weights)

BERT_CONFIGS = {
    "bert-tiny": dict(layers=2, hidden=128, heads=2, ffn=512),
    "bert-base": dict(layers=12, hidden=768, heads=12, ffn=3072),
    "bert-large": dict(layers=24, hidden=1024, heads=16, ffn=4096),
}


class BertCore(nn.Module):
    """Single-module BERT (used when num_stages == 1)."""

    def __init__(self, layers, hidden, heads, ffn, vocab_size, max_pos,
                 dropout=0.0):
        super().__init__()
        self.embeddings = Embeddings(vocab_size, hidden, max_pos)
        self.blocks = nn.ModuleList(
            Block(hidden, heads, ffn, causal=False, pre_ln=False,
                  dropout=dropout)
            for _ in range(layers))
        self.head = LMHead(hidden, vocab_size)

    def forward(self, ids):
        x = self.embeddings(ids)
        for b in self.blocks:
            x = b(x)
        return self.head(x)


class StagedModel(nn.Module):
    """Container for stage-chainable pipeline models."""

    def __init__(self, stages):
        super().__init__()
        self.stages = nn.ModuleList(stages)

    def forward(self, x):
        for s in self.stages:
            x = s(x)
        return x


class _Stage(nn.Module):
    def __init__(self, mods):
        super().__init__()
        self.mods = nn.ModuleList(mods)

    def forward(self, x):
        for m in self.mods:
            x = m(x)
        return x


def build_bert(config="bert-large", vocab_size=30528, max_pos=512,
               num_stages=1, dropout=0.0):
    """Build BERT under epl annotations.  vocab defaults to 30528
    (30522 rounded up to /8 for the fused CE kernel's vectorized path)."""
    cfg = BERT_CONFIGS[config] if isinstance(config, str) else dict(config)
    L, H, A, F = cfg["layers"], cfg["hidden"], cfg["heads"], cfg["ffn"]
    if num_stages <= 1:
        with epl.replicate(device_count=1, name="stage_0"):
            model = BertCore(L, H, A, F, vocab_size, max_pos,
                             dropout=dropout)
        return init_weights(model)
    per = (L + num_stages - 1) // num_stages
    stages = []
    layer_idx = 0
    for s in range(num_stages):
        with epl.replicate(device_count=1, name="stage_{}".format(s)):
            mods = []
            if s == 0:
                mods.append(Embeddings(vocab_size, H, max_pos))
            n = min(per, L - layer_idx)
            mods.extend(Block(H, A, F, dropout=dropout)
                        for _ in range(n))
            layer_idx += n
            if s == num_stages - 1:
                mods.append(LMHead(H, vocab_size))
            stages.append(_Stage(mods))
    return init_weights(StagedModel(stages))


def synthetic_mlm_batch(batch, seq_len, vocab_size=30528, device="cpu",
                        mask_frac=0.15, seed=None):
    g = torch.Generator(device="cpu")
    if seed is not None:
        g.manual_seed(seed)
    ids = torch.randint(0, vocab_size, (batch, seq_len), generator=g)
    targets = ids.clone()
    mask = torch.rand(batch, seq_len, generator=g) < mask_frac
    targets[~mask] = -100  # only predict masked positions
    ids[mask] = 103        # [MASK]
    return ids.to(device), targets.to(device)
