"""Parallel-strategy annotation base.

Capability parity: /root/reference/epl/strategies/parallel_strategy.py
(context-manager protocol :59-75, def-site identity :48-57).

MI355X redesign: a strategy scope is entered around *model construction*
code.  While active, every ``nn.Module`` constructed is tagged with the
strategy (see parallel/hooks.py); the engine later turns the ordered list of
strategies into taskgraphs (pipeline stages / split groups).  Identity is the
(name, definition call-site) pair, so re-entering the same annotation maps to
the same taskgraph — matching the reference's stack-hash identity.
"""

import traceback


class ParallelStrategy:
    def __init__(self, device_count=1, name=None):
        if device_count is None:
            device_count = 1
        if device_count < 1:
            raise ValueError("device_count must be >= 1")
        self.device_count = int(device_count)
        self.name = name
        # definition call-site identity (reference :48-57): file:line of the
        # frame that created this annotation, outside epl internals.
        stack = traceback.extract_stack()
        site = None
        for frame in reversed(stack):
            if "easyparallellibrary_amd" not in frame.filename:
                site = (frame.filename, frame.lineno)
                break
        self.def_site = site
        self.index = None  # assigned by StrategyContext

    @property
    def strategy_type(self):
        raise NotImplementedError

    @property
    def identity(self):
        if self.name is not None:
            return (self.strategy_type, self.name)
        return (self.strategy_type, self.def_site)

    def __enter__(self):
        from easyparallellibrary_amd.env import Env
        if Env.get().config.auto.auto_parallel:
            # reference: strategies/parallel_strategy.py:61-63
            raise ValueError("auto.auto_parallel is enabled; do not use "
                             "explicit strategy scopes at the same time")
        Env.get().strategy_context.add_context(self)
        return self

    def __exit__(self, exc_type, exc_val, exc_tb):
        from easyparallellibrary_amd.env import Env
        Env.get().strategy_context.remove_context(self)
        return False

    def __repr__(self):
        return "{}(device_count={}, name={!r}, index={})".format(
            type(self).__name__, self.device_count, self.name, self.index)
