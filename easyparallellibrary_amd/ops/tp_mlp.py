"""Megatron-style tensor-parallel transformer blocks (MLP + attention).

Beyond the reference's single-sharded-layer pattern: the FIRST linear is
column-parallel (output features sharded, NO gather), the activation runs
on the shard, and the SECOND linear is row-parallel over the pre-sharded
features, finishing with ONE all-reduce.  The pair's output is the FULL
tensor on every rank, which is exactly what pipeline stage boundaries
and residual streams need — this is the building block for dense-TP
pipeline stages (NOTES.md round-2 item 5).
"""

import torch.nn as nn

from easyparallellibrary_amd.comm import functional
from easyparallellibrary_amd.ops.distributed_dense import (
    ColumnParallelLinear, RowParallelLinear)


class TensorParallelMLP(nn.Module):
    """fc1 (column, sharded out) -> act -> fc2 (row, all-reduce out).

    Built with ``comm=None`` the layers stay full (plain Linears) and
    ``set_comm`` shards them later — the deferred pattern that lets the
    module live inside an ``epl.split`` scope (the engine's split
    transform calls set_comm with the scope's communicator, exactly as
    for ExpertParallelMLP)."""

    def __init__(self, hidden, ffn_hidden, comm=None, act=None,
                 source_fc1=None, source_fc2=None):
        super().__init__()
        self.hidden = hidden
        self.ffn_hidden = ffn_hidden
        self.comm = comm
        if comm is None or comm.size == 1:
            self.fc1 = source_fc1 or nn.Linear(hidden, ffn_hidden)
            self.fc2 = source_fc2 or nn.Linear(ffn_hidden, hidden)
        else:
            self.fc1 = ColumnParallelLinear(hidden, ffn_hidden, comm,
                                            bias=True, gather_input=False,
                                            source=source_fc1)
            self.fc2 = RowParallelLinear(ffn_hidden, hidden, comm,
                                         bias=True, source=source_fc2,
                                         pre_sharded=True)
        self.act = act if act is not None else nn.GELU()

    def set_comm(self, comm):
        """Shard the (still full) layers over ``comm`` in place."""
        if comm is None or comm.size == 1:
            self.comm = comm
            return
        assert isinstance(self.fc1, nn.Linear), "already sharded"
        self.comm = comm
        self.fc1 = ColumnParallelLinear(self.hidden, self.ffn_hidden,
                                        comm, bias=True,
                                        gather_input=False,
                                        source=self.fc1)
        self.fc2 = RowParallelLinear(self.ffn_hidden, self.hidden, comm,
                                     bias=True, source=self.fc2,
                                     pre_sharded=True)

    def forward(self, x):
        # Megatron 'f': identity forward, all-reduce of the partial
        # input-gradients backward
        x = functional.copy_to_group(x, self.comm)
        return self.fc2(self.act(self.fc1(x)))


class TensorParallelSelfAttention(nn.Module):
    """Head-sharded self-attention: q/k/v projections column-parallel
    (each rank owns heads/W full heads), attention runs on the local
    heads over the full sequence, the output projection is row-parallel
    — full tensor out after ONE all-reduce (the Megatron attention
    block).  Requires heads % comm.size == 0."""

    def __init__(self, hidden, num_heads, comm=None, causal=False,
                 source=None):
        super().__init__()
        assert hidden % num_heads == 0
        self.comm = comm
        self.hidden = hidden
        self.num_heads = num_heads
        self.head_dim = hidden // num_heads
        self.causal = causal
        sq = sk = sv = sp = None
        if source is not None:   # an ops-style module with q/k/v/proj
            sq, sk, sv, sp = (source.q, source.k, source.v, source.proj)
        w = comm.size if comm is not None else 1
        if w <= 1:
            self.local_heads = num_heads
            self.q = sq or nn.Linear(hidden, hidden)
            self.k = sk or nn.Linear(hidden, hidden)
            self.v = sv or nn.Linear(hidden, hidden)
            self.proj = sp or nn.Linear(hidden, hidden)
        else:
            self._shard(comm, sq, sk, sv, sp)

    def _shard(self, comm, sq, sk, sv, sp):
        w = comm.size
        assert self.num_heads % w == 0, "heads must divide TP degree"
        self.local_heads = self.num_heads // w
        hidden = self.hidden

        def col(src):
            return ColumnParallelLinear(hidden, hidden, comm, bias=True,
                                        gather_input=False, source=src)

        self.q = col(sq)
        self.k = col(sk)
        self.v = col(sv)
        self.proj = RowParallelLinear(hidden, hidden, comm, bias=True,
                                      source=sp, pre_sharded=True)

    def set_comm(self, comm):
        """Shard the (still full) projections over ``comm`` in place."""
        if comm is None or comm.size == 1:
            self.comm = comm
            return
        assert isinstance(self.q, nn.Linear), "already sharded"
        self.comm = comm
        self._shard(comm, self.q, self.k, self.v, self.proj)

    def forward(self, x):
        from easyparallellibrary_amd.ops.attention import flash_attention
        x = functional.copy_to_group(x, self.comm)
        b, s, _ = x.shape
        shp = (b, s, self.local_heads, self.head_dim)
        q = self.q(x).reshape(shp).transpose(1, 2)
        k = self.k(x).reshape(shp).transpose(1, 2)
        v = self.v(x).reshape(shp).transpose(1, 2)
        o = flash_attention(q, k, v, causal=self.causal)
        return self.proj(o.transpose(1, 2).reshape(b, s, -1))
