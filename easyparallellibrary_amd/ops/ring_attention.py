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

Each arriving K/V block is processed by the NATIVE flash kernels
(ops/attention.flash_attention_with_lse: the forward also returns the
per-row log-sum-exp as a differentiable output, and the backward folds
the incoming lse gradient into its per-row delta), and the partial
(out, lse) pairs merge with a streaming differentiable logaddexp —
autograd derives the backward through both the merge and the ring
exchanges (_RingShift: send-next/recv-prev, whose adjoint is the
opposite rotation).  Fully-masked causal blocks are skipped on the
compute side while the ring p2p stays in lockstep.  On CPU (or
non-bf16/d!=64,128) the same structure runs through the fp32 torch
fallback of flash_attention_with_lse, which is what the CPU exact-match
tests exercise.
"""

import torch
import torch.nn as nn

from easyparallellibrary_amd.ops.attention import flash_attention_with_lse


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


def ring_attention(q, k, v, comm, causal=False, scale=None):
    """q/k/v: [b, h, s_local, d] — each rank's sequence block.  Returns
    [b, h, s_local, d].  Requires equal block sizes on every rank.

    Per-block compute = flash_attention_with_lse (native CDNA4 kernels
    on GPU); partials merge with a streaming differentiable logaddexp:
      lse' = logaddexp(lse, lse_b)
      o'   = o * exp(lse - lse') + o_b * exp(lse_b - lse')
    Under causal masking, block src == rank runs the kernel's causal
    path (the diagonal aligns because block sizes are equal), earlier
    blocks run unmasked, and later (fully-masked) blocks are SKIPPED on
    the compute side — the ring_shift p2p still runs on every rank so
    the ring stays in lockstep."""
    if scale is None:
        scale = q.shape[-1] ** -0.5
    w = comm.size if comm is not None else 1
    rank = comm.rank if comm is not None else 0

    lse = o = dead = None
    kb, vb = k, v
    for step in range(w):
        src = (rank - step) % w            # whose K/V block we hold now
        if not (causal and src > rank):    # future blocks: skip compute
            ob, lseb = flash_attention_with_lse(
                q, kb, vb, causal=(causal and src == rank), scale=scale)
            ob = ob.float()
            if o is None:
                o, lse = ob, lseb
            else:
                lse_new = torch.logaddexp(lse, lseb)
                o = (o * (lse - lse_new).exp().unsqueeze(-1)
                     + ob * (lseb - lse_new).exp().unsqueeze(-1))
                lse = lse_new
        else:
            # zero-valued contribution keeps the skipped block in the
            # autograd graph, so every rank runs the SAME chain of
            # ring-shift backwards (the backward p2p must stay in
            # lockstep even for blocks that contribute nothing here)
            z = (kb.sum() + vb.sum()) * 0
            dead = z if dead is None else dead + z
        if step < w - 1:
            kb = ring_shift(kb, comm)
            vb = ring_shift(vb, comm)
    out = o.to(q.dtype)
    if dead is not None:
        out = out + dead.to(out.dtype)
    return out


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
