"""Plan IR: taskgraphs over the module tree.

Capability parity: /root/reference/epl/ir/graph.py (op->taskgraph
classification :518-569) and ir/taskgraph.py (Taskgraph :107).

MI355X redesign: the reference classifies every captured TF op into a
taskgraph at graph-build time.  Here the IR is a *plan over modules*: each
``nn.Module`` constructed under a ``replicate``/``split`` annotation carries
its strategy tag (parallel/hooks.py); ``Plan.build`` walks the module tree
once, groups the maximal tagged subtree roots into TaskGraphs in annotation
order, and assigns rank groups via the cluster Layout.  No op cloning ever
happens — replicas exist because every rank instantiates its own modules,
and micro-batching is a runtime loop (parallel/pipeline.py).
"""

from easyparallellibrary_amd import constant
from easyparallellibrary_amd.parallel import hooks


class TaskGraph:
    """One annotation scope: a pipeline stage (replicate) or a sharded group
    (split).  Reference: epl/ir/taskgraph.py:107."""

    def __init__(self, index, strategy):
        self.index = index
        self.strategy = strategy
        self.module_names = []   # maximal tagged subtree roots, in order
        self.modules = []
        self.virtual_device = None
        # exclusive parameter list (innermost-tag assignment); when set it
        # overrides the recursive module walk — a replicate subtree that
        # CONTAINS a split subtree must not claim the split parameters
        self.params_exclusive = None

    @property
    def strategy_type(self):
        return self.strategy.strategy_type

    @property
    def is_split(self):
        return self.strategy_type == constant.SPLIT

    @property
    def replicated_io(self):
        """True when this stage's boundary activations are identical on
        every position (dense-TP stages) — required at mixed-width
        pipeline boundaries (parallel/pipeline.py)."""
        return getattr(self.strategy, "replicated_io", False)

    @property
    def device_count(self):
        return self.strategy.device_count

    def parameters(self):
        if self.params_exclusive is not None:
            for p in self.params_exclusive:
                yield p
            return
        for m in self.modules:
            for p in m.parameters():
                yield p

    def __repr__(self):
        return "TaskGraph(index={}, type={}, dc={}, modules={})".format(
            self.index, self.strategy_type, self.device_count,
            self.module_names)


class Plan:
    """The parallelization plan for one model."""

    def __init__(self, taskgraphs, default_strategy=None):
        self.taskgraphs = taskgraphs
        self.default_strategy = default_strategy

    @property
    def num_stages(self):
        return sum(1 for tg in self.taskgraphs
                   if tg.strategy_type == constant.REPLICATE)

    @property
    def has_split(self):
        return any(tg.is_split for tg in self.taskgraphs)

    @classmethod
    def build(cls, model, strategy_context):
        """Group maximal tagged subtree roots into taskgraphs.

        A module belongs to taskgraph T if it carries T's tag and its parent
        does not (maximal root).  Untagged top-level modules (and the root
        model's own direct parameters) go to the default strategy's
        taskgraph, or to taskgraph 0 if no default was set (mirrors the
        reference's phase/colocation fallback, ir/graph.py:354-465).
        """
        strategies = strategy_context.strategies
        taskgraphs = [TaskGraph(s.index, s) for s in strategies]

        def tg_of(strategy):
            return taskgraphs[strategy.index]

        default = strategy_context._default_strategy
        fallback_tg = tg_of(default) if default is not None else None

        def visit(module, name, parent_strategy):
            s = hooks.strategy_of(module)
            effective = s if s is not None else parent_strategy
            if effective is not None and effective is not parent_strategy:
                tg = tg_of(effective)
                tg.module_names.append(name)
                tg.modules.append(module)
            for child_name, child in module.named_children():
                full = "{}.{}".format(name, child_name) if name else child_name
                visit(child, full, effective)

        root_strategy = hooks.strategy_of(model)
        if root_strategy is not None:
            # whole model under one scope; still descend — a nested scope
            # (e.g. split expert weights under a default-replicate model)
            # forms its own taskgraph
            tg = tg_of(root_strategy)
            tg.module_names.append("")
            tg.modules.append(model)
            for child_name, child in model.named_children():
                visit(child, child_name, root_strategy)
        else:
            start = default
            for child_name, child in model.named_children():
                visit(child, child_name, start)

        # untagged leftovers: modules not captured above but holding params.
        if fallback_tg is None:
            nonempty = [tg for tg in taskgraphs if tg.modules]
            fallback_tg = nonempty[0] if nonempty else None
        covered = set()
        for tg in taskgraphs:
            covered.update(id(m) for m in tg.modules)

        def is_covered(module):
            return id(module) in covered

        uncovered = []
        stack = [(model, "", False)]
        while stack:
            mod, name, inside = stack.pop()
            inside = inside or is_covered(mod)
            if not inside and (name != "") and any(
                    True for _ in mod.parameters(recurse=False)):
                uncovered.append((name, mod))
            for cn, c in mod.named_children():
                full = "{}.{}".format(name, cn) if name else cn
                stack.append((c, full, inside))
        if uncovered:
            if fallback_tg is None:
                raise ValueError(
                    "Model has parameters outside any replicate/split scope "
                    "and no default strategy was set: {}".format(
                        [n for n, _ in uncovered]))
            for name, mod in sorted(uncovered):
                fallback_tg.module_names.append(name)
                fallback_tg.modules.append(mod)

        # exclusive parameter assignment: every parameter belongs to the
        # taskgraph of its module's INNERMOST tag (untagged -> default /
        # fallback taskgraph)
        assignment = {}
        eff_default = default if default is not None else (
            fallback_tg.strategy if fallback_tg is not None else None)

        def assign(module, strategy):
            s = hooks.strategy_of(module) or strategy
            if s is not None:
                for p in module.parameters(recurse=False):
                    assignment.setdefault(id(p), (p, s))
            for c in module.children():
                assign(c, s)

        assign(model, eff_default)
        per_tg = {}
        for p, s in assignment.values():
            per_tg.setdefault(s.index, []).append(p)
        for tg in taskgraphs:
            if tg.strategy.index in per_tg:
                tg.params_exclusive = per_tg[tg.strategy.index]

        taskgraphs = [tg for tg in taskgraphs if tg.modules]
        for i, tg in enumerate(taskgraphs):
            tg.index = i
        return cls(taskgraphs, default)

    def effective_device_counts(self, colocate_split_and_replicate=False):
        """Per-taskgraph devices-per-replica for the Layout.  With
        colocation (reference: config.py:170-171), a split taskgraph shares
        the devices of the replicate taskgraph of equal device_count, so it
        contributes 0 extra devices."""
        counts = []
        seen_replicate = {}
        for tg in self.taskgraphs:
            dc = tg.device_count
            if tg.is_split and colocate_split_and_replicate and \
                    seen_replicate.get(dc):
                counts.append(0)
            else:
                counts.append(dc)
                if not tg.is_split:
                    seen_replicate[dc] = True
        return counts

    def __repr__(self):
        return "Plan({})".format(self.taskgraphs)
