"""Stack of active strategy annotations with the reference's nesting rules.

Capability parity: /root/reference/epl/strategies/strategy_context.py
(nesting checks :34-54, strategy index assignment, identity :133-135,
default strategy :144-152).
"""

from easyparallellibrary_amd import constant


class StrategyContext:
    def __init__(self):
        self._stack = []
        # ordered unique strategies (the taskgraph list), keyed by identity
        self._registry = {}
        self._ordered = []
        self._default_strategy = None

    # ---- registration -------------------------------------------------------
    def _register(self, strategy):
        key = strategy.identity
        if key in self._registry:
            canonical = self._registry[key]
            strategy.index = canonical.index
            return canonical
        strategy.index = len(self._ordered)
        self._registry[key] = strategy
        self._ordered.append(strategy)
        return strategy

    def add_context(self, strategy):
        for active in self._stack:
            if active.strategy_type == strategy.strategy_type:
                raise ValueError(
                    "Nesting two {} scopes is not allowed".format(
                        strategy.strategy_type))
            if active.strategy_type == constant.SPLIT:
                raise ValueError("No strategy may nest inside a split scope")
            if (active.strategy_type == constant.REPLICATE
                    and strategy.strategy_type == constant.SPLIT):
                raise ValueError(
                    "A split scope may not nest inside a replicate scope")
        canonical = self._register(strategy)
        self._stack.append(canonical)

    def remove_context(self, strategy):
        key = strategy.identity
        for i in range(len(self._stack) - 1, -1, -1):
            if self._stack[i].identity == key:
                del self._stack[i]
                return
        raise ValueError("Strategy {} not in context stack".format(strategy))

    # ---- queries ------------------------------------------------------------
    @property
    def current(self):
        if self._stack:
            return self._stack[-1]
        return self._default_strategy

    @property
    def strategies(self):
        return list(self._ordered)

    @property
    def num_taskgraphs(self):
        return len(self._ordered)

    def set_default_strategy(self, strategy):
        """reference: epl.set_default_strategy (strategy_context.py:144-152)"""
        self._default_strategy = self._register(strategy)

    def reset(self):
        self._stack = []
        self._registry = {}
        self._ordered = []
        self._default_strategy = None
