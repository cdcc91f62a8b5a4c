"""GPT-2 model family (BASELINE config 5: GPT-2 XL hybrid DPxPP + auto
gradient-checkpoint + CPU offload)."""

import torch
import torch.nn as nn

import easyparallellibrary_amd as epl
from easyparallellibrary_amd.models.bert import StagedModel, _Stage
from easyparallellibrary_amd.models.transformer import (Block, Embeddings,
                                                        LMHead, init_weights)

GPT2_CONFIGS = {
    "gpt2-tiny": dict(layers=2, hidden=128, heads=2, ffn=512),
    "gpt2-small": dict(layers=12, hidden=768, heads=12, ffn=3072),
    "gpt2-medium": dict(layers=24, hidden=1024, heads=16, ffn=4096),
    "gpt2-large": dict(layers=36, hidden=1280, heads=20, ffn=5120),
    "gpt2-xl": dict(layers=48, hidden=1600, heads=25, ffn=6400),
}


class GPT2Core(nn.Module):
    def __init__(self, layers, hidden, heads, ffn, vocab_size, max_pos,
                 dropout=0.0):
        super().__init__()
        self.embeddings = Embeddings(vocab_size, hidden, max_pos,
                                     use_ln=False)
        self.blocks = nn.ModuleList(
            Block(hidden, heads, ffn, causal=True, pre_ln=True,
                  dropout=dropout)
            for _ in range(layers))
        self.head = LMHead(hidden, vocab_size, use_ln=True)

    def forward(self, ids):
        x = self.embeddings(ids)
        for b in self.blocks:
            x = b(x)
        return self.head(x)


def build_gpt2(config="gpt2-xl", vocab_size=50264, max_pos=1024,
               num_stages=1, dropout=0.0):
    """vocab 50257 rounded up to /8 for the vectorized CE kernel."""
    cfg = GPT2_CONFIGS[config] if isinstance(config, str) else dict(config)
    L, H, A, F = cfg["layers"], cfg["hidden"], cfg["heads"], cfg["ffn"]
    if num_stages <= 1:
        with epl.replicate(device_count=1, name="stage_0"):
            model = GPT2Core(L, H, A, F, vocab_size, max_pos,
                             dropout=dropout)
        return init_weights(model)
    per = (L + num_stages - 1) // num_stages
    stages = []
    layer_idx = 0
    for s in range(num_stages):
        with epl.replicate(device_count=1, name="stage_{}".format(s)):
            mods = []
            if s == 0:
                mods.append(Embeddings(vocab_size, H, max_pos, use_ln=False))
            n = min(per, L - layer_idx)
            mods.extend(Block(H, A, F, causal=True, pre_ln=True,
                              dropout=dropout)
                        for _ in range(n))
            layer_idx += n
            if s == num_stages - 1:
                mods.append(LMHead(H, vocab_size, use_ln=True))
            stages.append(_Stage(mods))
    return init_weights(StagedModel(stages))


def synthetic_lm_batch(batch, seq_len, vocab_size=50264, device="cpu",
                       seed=None):
    g = torch.Generator(device="cpu")
    if seed is not None:
        g.manual_seed(seed)
    ids = torch.randint(0, vocab_size, (batch, seq_len + 1), generator=g)
    return ids[:, :-1].to(device), ids[:, 1:].contiguous().to(device)
