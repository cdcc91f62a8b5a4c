"""Data-parallel / pipeline-stage annotation.

Capability parity: /root/reference/epl/strategies/replicate.py:24-41.
``epl.replicate(device_count=1, name='stage_0')`` marks the modules built
inside the scope as one data-parallel taskgraph; several distinct replicate
scopes become pipeline stages.
"""

from easyparallellibrary_amd import constant
from easyparallellibrary_amd.strategies.base import ParallelStrategy


class Replicate(ParallelStrategy):
    @property
    def strategy_type(self):
        return constant.REPLICATE


def replicate(device_count=1, name=None):
    return Replicate(device_count=device_count, name=name)
