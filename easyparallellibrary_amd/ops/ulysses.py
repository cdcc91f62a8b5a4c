"""Ulysses-style sequence parallelism (DeepSpeed-Ulysses pattern).

Beyond-parity extension: the reference has NO sequence/context parallelism
(SURVEY.md section 5 — its building blocks are the MoE all-to-all and the
allgather bridges).  This module builds head-scatter / seq-gather
attention on exactly those primitives: each rank holds seq/W tokens of
every head before attention; one all-to-all turns that into all tokens of
heads/W, attention runs over the full sequence on the local heads, and
the reverse all-to-all restores the sequence sharding.  The a2a autograd
pair (comm/functional.py, a2a <-> a2a) makes the backward automatic, and
on GPU both exchanges ride RCCL over xGMI.
"""

import torch
import torch.nn as nn

from easyparallellibrary_amd.comm import functional
from easyparallellibrary_amd.ops.attention import flash_attention


def seq_to_heads(x, comm):
    """[b, s_local, H, d] -> [b, s_full, H/W, d] (scatter heads, gather
    sequence)."""
    if comm is None or comm.size == 1:
        return x
    w = comm.size
    b, sl, H, d = x.shape
    assert H % w == 0, "heads must divide the sequence-parallel degree"
    hl = H // w
    # chunk w of dim0 = head-group w of my local tokens
    t = x.reshape(b, sl, w, hl, d).permute(2, 0, 1, 3, 4).contiguous()
    t = functional.all_to_all(t, comm)          # [w, b, sl, hl, d]
    # received chunk w = my head-group of rank w's tokens = seq chunk w
    return t.permute(1, 0, 2, 3, 4).reshape(b, w * sl, hl, d)


def heads_to_seq(x, comm):
    """[b, s_full, H/W, d] -> [b, s_local, H, d] (inverse exchange)."""
    if comm is None or comm.size == 1:
        return x
    w = comm.size
    b, s, hl, d = x.shape
    sl = s // w
    t = x.reshape(b, w, sl, hl, d).permute(1, 0, 2, 3, 4).contiguous()
    t = functional.all_to_all(t, comm)          # [w, b, sl, hl, d]
    return t.permute(1, 2, 0, 3, 4).reshape(b, sl, w * hl, d)


class UlyssesSelfAttention(nn.Module):
    """Drop-in for SelfAttention under sequence sharding: the module takes
    the rank-local [b, s/W, hidden] slice and returns the same shape; the
    attention itself sees the full sequence."""

    def __init__(self, hidden, num_heads, comm=None, causal=False):
        super().__init__()
        assert hidden % num_heads == 0
        self.hidden = hidden
        self.num_heads = num_heads
        self.head_dim = hidden // num_heads
        self.causal = causal
        self.comm = comm
        self.qkv = nn.Linear(hidden, 3 * hidden)
        self.proj = nn.Linear(hidden, hidden)

    def set_comm(self, comm):
        self.comm = comm

    def forward(self, x):
        b, sl, h = x.shape
        qkv = self.qkv(x).reshape(b, sl, 3, self.num_heads, self.head_dim)
        q, k, v = qkv.unbind(dim=2)             # [b, sl, H, d]
        q = seq_to_heads(q, self.comm)          # [b, s, H/W, d]
        k = seq_to_heads(k, self.comm)
        v = seq_to_heads(v, self.comm)
        q = q.transpose(1, 2)
        k = k.transpose(1, 2)
        v = v.transpose(1, 2)
        o = flash_attention(q, k, v, causal=self.causal)
        o = o.transpose(1, 2)                   # [b, s, H/W, d]
        o = heads_to_seq(o, self.comm)          # [b, sl, H, d]
        return self.proj(o.reshape(b, sl, h))
