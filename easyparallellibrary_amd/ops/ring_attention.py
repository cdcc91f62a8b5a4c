"""Ring attention (context parallelism over the sequence dimension).

Beyond-parity extension #2 (SURVEY.md section 5 lists ring/context
parallelism as absent from the reference; Ulysses — ops/ulysses.py — is
extension #1).  Where Ulysses exchanges heads<->sequence with one
all-to-all (and therefore needs heads >= world), ring attention keeps
EVERY head local and instead rotates K/V blocks around the ring: rank r
holds queries for its sequence block and accumulates online-softmax
partials against each K/V block as it arrives.  W-1 point-to-point
hops overlap naturally with the partial-attention compute, and the
sequence length per rank is bounded only by memory.

The block math is the same online softmax the flash kernels use
(running max m, running sum l, rescaled accumulator o); here it is
expressed in differentiable torch ops so autograd derives the backward
through the ring exchanges (_RingShift: send-next/recv-prev, whose
adjoint is the opposite rotation).  On GPU the per-block matmuls ride
hipBLASLt and the exchanges ride RCCL over xGMI; fusing the block loop
into the flash kernels is queued for a later round.
"""

import torch
import torch.nn as nn


class _RingShift(torch.autograd.Function):
    """Circular rotation: send my block to rank+1, receive rank-1's.
    The adjoint rotates the other way."""

    @staticmethod
    def forward(ctx, x, comm):
        ctx.comm = comm
        w, r = comm.size, comm.rank
        x = x.contiguous()
        out = torch.empty_like(x, memory_format=torch.contiguous_format)
        comm.batch_p2p([(True, x, (r + 1) % w),
                        (False, out, (r - 1) % w)])
        return out

    @staticmethod
    def backward(ctx, g):
        comm = ctx.comm
        w, r = comm.size, comm.rank
        g = g.contiguous()
        out = torch.empty_like(g, memory_format=torch.contiguous_format)
        comm.batch_p2p([(True, g, (r - 1) % w),
                        (False, out, (r + 1) % w)])
        return out, None


def ring_shift(x, comm):
    if comm is None or comm.size == 1:
        return x
    return _RingShift.apply(x, comm)


def _block_attention(q, k, v, scale, mask):
    """Partial attention of q against one K/V block; returns the
    un-normalized (m, l, o) online-softmax triple (all fp32)."""
    s = (q.float() @ k.float().transpose(-1, -2)) * scale
    if mask is not None:
        s = s.masked_fill(mask, float("-inf"))
    m = s.amax(dim=-1, keepdim=True)
    # fully masked rows: keep m finite so exp() stays 0 without NaN
    m = torch.where(torch.isfinite(m), m, torch.zeros_like(m))
    p = (s - m).exp()
    l = p.sum(dim=-1, keepdim=True)
    o = p @ v.float()
    return m, l, o


def ring_attention(q, k, v, comm, causal=False, scale=None):
    """q/k/v: [b, h, s_local, d] — each rank's sequence block.  Returns
    [b, h, s_local, d].  Requires equal block sizes on every rank."""
    if scale is None:
        scale = q.shape[-1] ** -0.5
    w = comm.size if comm is not None else 1
    rank = comm.rank if comm is not None else 0
    sl = q.shape[-2]

    m = l = o = None
    kb, vb = k, v
    for step in range(w):
        src = (rank - step) % w            # whose K/V block we hold now
        mask = None
        if causal:
            qi = torch.arange(rank * sl, (rank + 1) * sl,
                              device=q.device).unsqueeze(-1)
            ki = torch.arange(src * sl, (src + 1) * sl,
                              device=q.device).unsqueeze(0)
            mask = ki > qi                  # future keys masked
            # fully masked blocks still run (zero contribution) so the
            # ring stays in lockstep on every rank
        mb, lb, ob = _block_attention(q, kb, vb, scale, mask)
        if m is None:
            m, l, o = mb, lb, ob
        else:
            m_new = torch.maximum(m, mb)
            a = (m - m_new).exp()
            b_ = (mb - m_new).exp()
            l = l * a + lb * b_
            o = o * a + ob * b_
            m = m_new
        if step < w - 1:
            kb = ring_shift(kb, comm)
            vb = ring_shift(vb, comm)
    return (o / l.clamp_min(1e-20)).to(q.dtype)


class RingSelfAttention(nn.Module):
    """Drop-in self-attention under sequence sharding: takes the
    rank-local [b, s/W, hidden] slice, returns the same shape.  Unlike
    Ulysses this keeps all heads on every rank (works for heads < W) and
    moves K/V blocks instead."""

    def __init__(self, hidden, num_heads, comm=None, causal=False):
        super().__init__()
        assert hidden % num_heads == 0
        self.hidden = hidden
        self.num_heads = num_heads
        self.head_dim = hidden // num_heads
        self.comm = comm
        self.causal = causal
        self.qkv = nn.Linear(hidden, 3 * hidden)
        self.proj = nn.Linear(hidden, hidden)

    def forward(self, x):
        b, s, h = x.shape
        qkv = self.qkv(x).reshape(b, s, 3, self.num_heads, self.head_dim)
        q = qkv[:, :, 0].transpose(1, 2)
        k = qkv[:, :, 1].transpose(1, 2)
        v = qkv[:, :, 2].transpose(1, 2)
        o = ring_attention(q, k, v, self.comm, causal=self.causal)
        return self.proj(o.transpose(1, 2).reshape(b, s, h))
