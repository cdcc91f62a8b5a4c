"""Column-parallel (sharded) linear layer.

Capability parity: /root/reference/epl/ops/distributed_dense.py — per-shard
kernel ``[in, units/n (+remainder on shard 0)]`` (:99-126), inputs gathered
across replicas first via Replica2Split (:127-137), checkpoint layout
``kernel_<shard>`` / ``bias_<shard>`` (:111-123).

MI355X note: the shard matmul itself is a plain library GEMM (hipBLASLt via
torch.nn.functional.linear); the collectives around it ride the engine's
TP communicator over xGMI.
"""

import torch
import torch.nn as nn
import torch.nn.functional as F

from easyparallellibrary_amd.ops import bridging
from easyparallellibrary_amd.ops.initializers import (
    distributed_glorot_uniform_)


def shard_size(total, nshards, shard):
    """Reference remainder policy (:102-108): remainder goes to shard 0."""
    base = total // nshards
    rem = total % nshards
    return base + (rem if shard == 0 else 0)


def shard_offset(total, nshards, shard):
    base = total // nshards
    rem = total % nshards
    if shard == 0:
        return 0
    return rem + shard * base


class ColumnParallelLinear(nn.Module):
    """Output-feature-sharded linear.  Forward: gather the batch across the
    TP group (Replica2Split bridge), local GEMM on the shard, producing the
    local slice of the output features for the FULL batch."""

    def __init__(self, in_features, out_features, comm, bias=True,
                 gather_input=True, source=None):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.comm = comm
        self.nshards = comm.size
        self.shard = max(comm.rank, 0)
        self.gather_input = gather_input
        self.local_out = shard_size(out_features, self.nshards, self.shard)
        self.offset = shard_offset(out_features, self.nshards, self.shard)
        self.weight = nn.Parameter(
            torch.empty(self.local_out, in_features))
        self.weight._epl_shard_dim = 0
        self.bias = nn.Parameter(torch.zeros(self.local_out)) if bias \
            else None
        if self.bias is not None:
            self.bias._epl_shard_dim = 0
        if source is not None:
            with torch.no_grad():
                self.weight.copy_(
                    source.weight[self.offset:self.offset + self.local_out])
                if bias and source.bias is not None:
                    self.bias.copy_(
                        source.bias[self.offset:self.offset + self.local_out])
        else:
            distributed_glorot_uniform_(self.weight, in_features,
                                        out_features)

    def forward(self, x):
        if self.gather_input:
            x = bridging.replica_to_split(x, self.comm)
        return F.linear(x, self.weight, self.bias)

    def extra_repr(self):
        return "in={}, out={}/{} (shard {}/{})".format(
            self.in_features, self.local_out, self.out_features, self.shard,
            self.nshards)


class RowParallelLinear(nn.Module):
    """Input-feature-sharded linear: local GEMM then allreduce of the
    partial outputs (the TP all-reduce over xGMI of BASELINE config 4)."""

    def __init__(self, in_features, out_features, comm, bias=True,
                 source=None, pre_sharded=False):
        super().__init__()
        from easyparallellibrary_amd.comm import functional
        self._fn = functional
        # pre_sharded: the input already IS this rank's in-feature shard
        # (e.g. it came from a paired ColumnParallelLinear) — skip the
        # local slice
        self.pre_sharded = pre_sharded
        self.comm = comm
        self.nshards = comm.size
        self.shard = max(comm.rank, 0)
        self.local_in = shard_size(in_features, self.nshards, self.shard)
        self.in_offset = shard_offset(in_features, self.nshards, self.shard)
        self.in_features = in_features
        self.out_features = out_features
        self.weight = nn.Parameter(torch.empty(out_features, self.local_in))
        self.weight._epl_shard_dim = 1
        self.bias = nn.Parameter(torch.zeros(out_features)) if bias else None
        if source is not None:
            with torch.no_grad():
                self.weight.copy_(source.weight[
                    :, self.in_offset:self.in_offset + self.local_in])
                if bias and source.bias is not None:
                    self.bias.copy_(source.bias)
        else:
            distributed_glorot_uniform_(self.weight, in_features,
                                        out_features)

    def forward(self, x):
        xs = (x if self.pre_sharded
              else x[..., self.in_offset:self.in_offset + self.local_in])
        out = F.linear(xs, self.weight)
        # Megatron 'g': sum the partials forward, pass the (replicated)
        # output-gradient straight through backward
        out = self._fn.reduce_from_group(out, self.comm)
        if self.bias is not None:
            out = out + self.bias
        return out
