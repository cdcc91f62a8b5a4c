"""Tensor/expert model-parallel annotation.

Capability parity: /root/reference/epl/strategies/split.py:24-51 (is_nested
guard :36-46).  Modules built inside a ``split`` scope are sharded across
``device_count`` ranks: Linear layers become column-parallel
(ops/distributed_dense.py), losses become sharded-vocab, expert weights get
MoE all-to-all dispatch (ops/moe.py).
"""

from easyparallellibrary_amd import constant
from easyparallellibrary_amd.strategies.base import ParallelStrategy


class Split(ParallelStrategy):
    @property
    def strategy_type(self):
        return constant.SPLIT


def split(device_count=1, name=None):
    return Split(device_count=device_count, name=name)
