"""Static per-module flops / bytes cost model.

Capability parity: /root/reference/epl/profiler/profiler.py:36-60
(profile_flops / profile_memory — static per-op flops and per-tensor
bytes feeding auto-GC and auto-stage) and flops.py's extra registrations
(:47-117).  Here the units are modules: Linear/Conv/attention flops from
their shapes, memory from parameter + activation estimates.
"""

import torch.nn as nn


def module_flops(mod, seq_len=512, batch=1):
    """Forward flops estimate for one sample of the given shape."""
    total = 0
    for m in mod.modules():
        if isinstance(m, nn.Linear):
            total += 2 * m.in_features * m.out_features * seq_len * batch
        elif isinstance(m, nn.Conv2d):
            k = m.kernel_size[0] * m.kernel_size[1]
            total += 2 * m.in_channels * m.out_channels * k * 56 * 56
        elif isinstance(m, nn.Embedding):
            total += 0
        else:
            w = sum(p.numel() for p in m.parameters(recurse=False))
            total += 2 * w * seq_len * batch
    return total


def profile_flops(model, seq_len=512, batch=1):
    """Per-top-level-child forward flops."""
    return {name: module_flops(m, seq_len, batch)
            for name, m in model.named_children()}


def profile_memory(model, dtype_bytes=2, seq_len=512, batch=1):
    """Per-top-level-child bytes: parameters + rough activation estimate."""
    out = {}
    for name, m in model.named_children():
        params = sum(p.numel() for p in m.parameters()) * dtype_bytes
        # activation estimate: output features x tokens
        act = 0
        for mm in m.modules():
            if isinstance(mm, nn.Linear):
                act += mm.out_features * seq_len * batch * dtype_bytes
        out[name] = params + act
    return out
