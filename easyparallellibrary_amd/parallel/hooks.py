"""Interception layer: tag every constructed nn.Module with the active
strategy annotation.

Capability parity: /root/reference/epl/parallel/hooks.py (add_hooks
:1000-1056; op->taskgraph classification happens in ir/graph.py:518-569).

MI355X redesign: the reference monkey-patches ~25 TF graph internals to
capture ops as they are created.  In the PyTorch module/autograd world the
only hook needed at definition time is on ``nn.Module.__init__``: while a
``replicate``/``split`` scope is active, each module constructed inside it
records the scope's taskgraph index.  All graph transformation happens later
in the engine (parallel/engine.py) on the *module tree*, not on captured ops
— replica "cloning" is per-rank instantiation, so the O(ops x replicas x
micro_batches) graph surgery of the reference (graph_editor.py) disappears.
"""

import functools

import torch.nn as nn

_TAG = "_epl_strategy"
_orig_module_init = None


def _tagging_init(orig_init):
    @functools.wraps(orig_init)
    def wrapped(self, *args, **kwargs):
        from easyparallellibrary_amd.env import Env
        orig_init(self, *args, **kwargs)
        strategy = Env.get().strategy_context.current
        if strategy is not None and not hasattr(self, _TAG):
            object.__setattr__(self, _TAG, strategy)
    return wrapped


def add_hooks():
    """Install the module-construction hook (idempotent)."""
    global _orig_module_init
    if _orig_module_init is not None:
        return
    _orig_module_init = nn.Module.__init__
    nn.Module.__init__ = _tagging_init(_orig_module_init)


def remove_hooks():
    global _orig_module_init
    if _orig_module_init is not None:
        nn.Module.__init__ = _orig_module_init
        _orig_module_init = None


def strategy_of(module):
    """The strategy annotation a module was constructed under (or None)."""
    return getattr(module, _TAG, None)


def set_strategy(module, strategy):
    object.__setattr__(module, _TAG, strategy)
