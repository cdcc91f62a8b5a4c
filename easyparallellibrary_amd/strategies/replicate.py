"""Data-parallel / pipeline-stage annotation.

Capability parity: /root/reference/epl/strategies/replicate.py:24-41.
``epl.replicate(device_count=1, name='stage_0')`` marks the modules built
inside the scope as one data-parallel taskgraph; several distinct replicate
scopes become pipeline stages.
"""

from easyparallellibrary_amd import constant
from easyparallellibrary_amd.strategies.base import ParallelStrategy


class Replicate(ParallelStrategy):
    def __init__(self, device_count=1, name=None, replicated_io=False):
        super().__init__(device_count=device_count, name=name)
        # Mixed-width pipelines (stages with different device counts) are
        # only well-defined when the wide stage's input/output activations
        # are IDENTICAL on every one of its positions (dense Megatron-TP
        # stages, where copy_to_group/all-reduce keep tensors replicated).
        # The stage must declare that explicitly — the runtime refuses to
        # guess from module types (NOTES.md mixed-width design sketch).
        self.replicated_io = bool(replicated_io)

    @property
    def strategy_type(self):
        return constant.REPLICATE


def replicate(device_count=1, name=None, replicated_io=False):
    return Replicate(device_count=device_count, name=name,
                     replicated_io=replicated_io)
