"""Auto-stage planner: partition an un-staged model into pipeline stages.

Capability parity: /root/reference/epl/parallel/planner.py
(AutoStageGenerator.search :49-112 — policy: repeated-block structure
first, fall back to balanced partition).

MI355X redesign: the unit list is the model's ordered leaf-chain — the
children of its single sequential spine (nn.Sequential / ModuleList
container) — instead of captured TF ops.  Stage weights are parameter
counts plus a flops-proportional term from the static cost model
(profiler/cost_model.py).  The result feeds the same pipeline runtime as
hand-annotated stages.
"""

import torch.nn as nn

from easyparallellibrary_amd.parallel.partitioner import partition_balance
from easyparallellibrary_amd.profiler.cost_model import module_flops


class AutoStageGenerator:
    def __init__(self, model, num_stages):
        self.model = model
        self.num_stages = num_stages

    def _find_spine(self):
        """The longest sequential chain of child modules."""
        best = None
        for mod in self.model.modules():
            if isinstance(mod, (nn.Sequential, nn.ModuleList)):
                if best is None or len(mod) > len(best):
                    best = mod
        return best

    def search(self):
        """Returns a list of num_stages lists of modules (the chain split),
        or None if the model has no usable sequential spine."""
        spine = self._find_spine()
        if spine is None or len(spine) < self.num_stages:
            return None
        items = list(spine)
        weights = []
        for m in items:
            p = sum(x.numel() for x in m.parameters())
            f = module_flops(m)
            weights.append(p + f / 1e3 + 1)
        parts = partition_balance(weights, self.num_stages)
        return [[items[i] for i in part] for part in parts]
