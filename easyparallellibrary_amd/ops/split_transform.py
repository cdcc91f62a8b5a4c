"""Split-scope module transformation.

Capability parity: /root/reference/epl/parallel/hooks.py split-mode op
replacement — dense :710-723 (-> ops/distributed_dense.py), add_weight
row-sharding :667-707, sparse softmax :726, argmax :742, equal :797.

MI355X redesign: the reference swaps TF python calls at trace time; here
the engine rewrites the already-constructed module tree of a split
taskgraph once at plan time: nn.Linear -> ColumnParallelLinear (weight
shard copied out of the full layer so initialization statistics match
single-device training exactly), nn.Embedding -> row-sharded embedding.
The model's forward code is untouched.
"""

import torch
import torch.nn as nn

from easyparallellibrary_amd.comm import functional
from easyparallellibrary_amd.ops.distributed_dense import (
    ColumnParallelLinear, shard_offset, shard_size)
from easyparallellibrary_amd.utils.logging import get_logger

logger = get_logger()


class VocabParallelEmbedding(nn.Module):
    """Row-sharded embedding (reference add_weight dim-0 sharding,
    hooks.py:667-707): each shard holds rows [begin, begin+local); out-of-
    range ids embed to zero and the partial outputs are all-reduced."""

    def __init__(self, num_embeddings, embedding_dim, comm, source=None):
        super().__init__()
        self.comm = comm
        self.nshards = comm.size
        self.shard = max(comm.rank, 0)
        self.num_embeddings = num_embeddings
        self.local_rows = shard_size(num_embeddings, self.nshards, self.shard)
        self.begin = shard_offset(num_embeddings, self.nshards, self.shard)
        self.weight = nn.Parameter(
            torch.empty(self.local_rows, embedding_dim))
        self.weight._epl_shard_dim = 0
        if source is not None:
            with torch.no_grad():
                self.weight.copy_(
                    source.weight[self.begin:self.begin + self.local_rows])
        else:
            nn.init.normal_(self.weight)

    def forward(self, ids):
        in_range = (ids >= self.begin) & (ids < self.begin + self.local_rows)
        safe = (ids - self.begin).clamp(0, self.local_rows - 1)
        out = nn.functional.embedding(safe, self.weight)
        out = out * in_range.unsqueeze(-1).to(out.dtype)
        return functional.all_reduce(out, self.comm)


def _replace_module(parent, name, new):
    setattr(parent, name, new)


# Between a paired column and row linear the features are SHARDED, so only
# feature-independent elementwise modules may sit in between (LayerNorm and
# friends normalise over the full feature dim and would be wrong).
_ELEMENTWISE_SAFE = (nn.ReLU, nn.GELU, nn.SiLU, nn.Tanh, nn.Sigmoid,
                     nn.LeakyReLU, nn.ELU, nn.Hardswish, nn.Mish,
                     nn.Dropout, nn.Identity)


class PairedColumnLinear(nn.Module):
    """Entry half of an auto-paired Megatron block: the 'f' copy operator
    (identity forward, all-reduce of input grads backward) in front of a
    column-parallel linear, so the pair's backward matches ops/tp_mlp.

    The column linear is NOT a registered submodule — its parameters are
    registered directly on this wrapper under the ORIGINAL Linear's
    names (weight/bias), so state-dict keys are identical with
    auto_pair_sequential on or off and checkpoints stay portable across
    the flag (advisor finding r1)."""

    def __init__(self, col, comm):
        super().__init__()
        self.__dict__["col"] = col   # plain attr: keep it out of the tree
        self.comm = comm
        self.weight = col.weight
        if col.bias is not None:
            self.bias = col.bias

    def forward(self, x):
        return self.col(functional.copy_to_group(x, self.comm))


def _pair_sequential(seq, comm):
    """Greedy left-to-right pairing of Linear..Linear runs inside an
    ``nn.Sequential`` whose in-between modules are all elementwise-safe:
    first Linear -> column-parallel (sharded out, no gather), second ->
    row-parallel over the pre-sharded features (ONE all-reduce instead of
    two all-gathers).  Sequential-only because there the execution order
    is known from structure — pairing arbitrary sibling Linears would be
    guessing at the user's forward.  Returns ids of consumed modules."""
    from easyparallellibrary_amd.ops.distributed_dense import (
        RowParallelLinear)
    children = list(seq.named_children())
    consumed = set()
    i = 0
    while i < len(children):
        name_i, mod_i = children[i]
        if not isinstance(mod_i, nn.Linear) or id(mod_i) in consumed:
            i += 1
            continue
        j = i + 1
        mod_j = None
        while j < len(children):
            cand = children[j][1]
            if isinstance(cand, nn.Linear):
                mod_j = cand
                break
            if not isinstance(cand, _ELEMENTWISE_SAFE):
                break
            j += 1
        if mod_j is None or mod_j.in_features != mod_i.out_features:
            i += 1
            continue
        name_j = children[j][0]
        col = ColumnParallelLinear(mod_i.in_features, mod_i.out_features,
                                   comm, bias=mod_i.bias is not None,
                                   gather_input=False, source=mod_i)
        row = RowParallelLinear(mod_j.in_features, mod_j.out_features,
                                comm, bias=mod_j.bias is not None,
                                source=mod_j, pre_sharded=True)
        setattr(seq, name_i, PairedColumnLinear(col, comm))
        setattr(seq, name_j, row)
        consumed.add(id(mod_i))
        consumed.add(id(mod_j))
        logger.info("auto-paired Sequential linears [%s] -> column / "
                    "[%s] -> row (%d-way)", name_i, name_j, comm.size)
        i = j + 1
    return consumed


def transform_taskgraph(tg, comm, gather_input=True, model=None):
    """Rewrite every Linear/Embedding under the taskgraph's module roots
    into its sharded equivalent over ``comm``.  ``model`` (the user's root
    module) lets a taskgraph root that is itself a Linear be rewired into
    its parent by dotted name."""
    from easyparallellibrary_amd.ops.moe import ExpertParallelMLP
    from easyparallellibrary_amd.ops.tp_mlp import (
        TensorParallelMLP, TensorParallelSelfAttention)
    deferred = (ExpertParallelMLP, TensorParallelMLP,
                TensorParallelSelfAttention)
    from easyparallellibrary_amd.env import Env
    auto_pair = (Env.get().config.auto.auto_pair_sequential
                 and comm.size > 1)
    replaced = 0
    done = set()  # module ids already transformed (named_modules visits
    #               a nested module both as child and as parent)
    for root_i, root in enumerate(tg.modules):
        if isinstance(root, (nn.Linear, nn.Embedding)):
            new = _make_sharded(root, comm, gather_input)
            tg.modules[root_i] = new
            if model is not None:
                name = tg.module_names[root_i]
                parts = name.split(".")
                parent = model
                for p in parts[:-1]:
                    parent = getattr(parent, p)
                setattr(parent, parts[-1], new)
            replaced += 1
            continue
        if isinstance(root, deferred):
            root.set_comm(comm)
            replaced += 1
            continue
        # materialize: set_comm / child replacement mutates the tree
        if auto_pair and isinstance(root, nn.Sequential):
            paired = _pair_sequential(root, comm)
            done |= paired
            replaced += len(paired)
        for parent_name, parent in list(root.named_modules()):
            if auto_pair and isinstance(parent, nn.Sequential) \
                    and parent is not root:
                paired = _pair_sequential(parent, comm)
                done |= paired
                replaced += len(paired)
            if isinstance(parent, deferred):
                if id(parent) not in done:
                    done.add(id(parent))
                    parent.set_comm(comm)
                    replaced += 1
                continue
            for child_name, child in list(parent.named_children()):
                if id(child) in done:
                    continue
                if isinstance(child, deferred):
                    done.add(id(child))
                    child.set_comm(comm)
                    replaced += 1
                elif isinstance(child, (nn.Linear, nn.Embedding)):
                    done.add(id(child))
                    _replace_module(parent, child_name,
                                    _make_sharded(child, comm, gather_input))
                    replaced += 1
    # parameters changed object identity: refresh the taskgraph's
    # exclusive param list from the (transformed) module roots
    tg.params_exclusive = None
    logger.info("split transform: %d module(s) sharded %d-way over %s",
                replaced, comm.size, comm.name)
    return replaced


def _make_sharded(mod, comm, gather_input):
    if isinstance(mod, nn.Linear):
        return ColumnParallelLinear(
            mod.in_features, mod.out_features, comm,
            bias=mod.bias is not None, gather_input=gather_input, source=mod)
    if isinstance(mod, nn.Embedding):
        return VocabParallelEmbedding(
            mod.num_embeddings, mod.embedding_dim, comm, source=mod)
    raise TypeError(type(mod))
