"""Megatron-style tensor-parallel MLP pair.

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
    """fc1 (column, sharded out) -> act -> fc2 (row, all-reduce out)."""

    def __init__(self, hidden, ffn_hidden, comm, act=None, source_fc1=None,
                 source_fc2=None):
        super().__init__()
        self.fc1 = ColumnParallelLinear(hidden, ffn_hidden, comm,
                                        bias=True, gather_input=False,
                                        source=source_fc1)
        self.act = act if act is not None else nn.GELU()
        self.fc2 = RowParallelLinear(ffn_hidden, hidden, comm, bias=True,
                                     source=source_fc2, pre_sharded=True)

    def forward(self, x):
        # Megatron 'f': identity forward, all-reduce of the partial
        # input-gradients backward
        x = functional.copy_to_group(x, self.fc1.comm)
        return self.fc2(self.act(self.fc1(x)))
