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

from easyparallellibrary_amd.parallel.partitioner import (
    find_repeated_blocks, partition_balance)
from easyparallellibrary_amd.profiler.cost_model import module_flops
from easyparallellibrary_amd.utils.logging import get_logger

logger = get_logger()


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

    def _weights(self, items):
        out = []
        for m in items:
            p = sum(x.numel() for x in m.parameters())
            f = module_flops(m)
            out.append(p + f / 1e3 + 1)
        return out

    def search(self):
        """Returns a list of num_stages lists of modules (the chain
        split), or None if the model has no usable sequential spine.

        Policy order matches the reference (planner.py:66-112): cut at
        REPEATED-BLOCK boundaries first (the transformer spine's dominant
        repeated module class), weight-balanced cuts only as fallback.
        With block cuts, the non-repeated prefix (embeddings) joins the
        first stage and the suffix (head) the last, and stage loads are
        balanced over whole blocks by the same DP used for the fallback.
        """
        spine = self._find_spine()
        if spine is None or len(spine) < self.num_stages:
            return None
        items = list(spine)
        weights = self._weights(items)
        k = self.num_stages

        block_key = find_repeated_blocks(items, min_repeat=max(3, k))
        if block_key is not None:
            starts = [i for i, m in enumerate(items)
                      if type(m).__name__ == block_key]
            if len(starts) >= k:
                # segment s = block s plus everything up to the next
                # block; the prefix before block 0 joins segment 0
                seg_bounds = [0] + starts[1:] + [len(items)]
                seg_w = [sum(weights[seg_bounds[s]:seg_bounds[s + 1]])
                         for s in range(len(starts))]
                parts = partition_balance(seg_w, k)
                out = []
                for part in parts:
                    lo = seg_bounds[part[0]]
                    hi = seg_bounds[part[-1] + 1]
                    out.append(items[lo:hi])
                logger.info(
                    "auto-stage: repeated-block policy on %d x %s "
                    "(cuts at block boundaries)", len(starts), block_key)
                return out

        parts = partition_balance(weights, k)
        return [[items[i] for i in part] for part in parts]
