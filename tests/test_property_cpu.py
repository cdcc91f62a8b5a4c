"""Property-based tests (hypothesis) for the pure-logic layers: the
weighted partitioner, the shard-size policy, IO slicing, and the bucket
layout — invariants the example-based tests can't sweep."""

import hypothesis.strategies as st
import torch
from hypothesis import given, settings

from easyparallellibrary_amd.ops.distributed_dense import (shard_offset,
                                                           shard_size)
from easyparallellibrary_amd.parallel.partitioner import (
    find_repeated_blocks, partition_balance)
from easyparallellibrary_amd.utils.io_slicing import slice_files


@settings(max_examples=200, deadline=None)
@given(st.lists(st.integers(min_value=0, max_value=10 ** 6), min_size=1,
                max_size=40),
       st.integers(min_value=1, max_value=12))
def test_partition_balance_invariants(weights, k):
    parts = partition_balance(weights, k)
    # contiguous, ordered, exact cover
    flat = [i for part in parts for i in part]
    assert flat == list(range(len(weights)))
    assert len(parts) == min(k, len(weights))
    # DP optimality vs every contiguous 2-split refinement: the max
    # chunk cost can never beat the trivial lower bounds
    costs = [sum(weights[i] for i in part) for part in parts]
    assert max(costs) >= max(weights)                # one item per chunk min
    assert max(costs) >= sum(weights) / len(parts) - 1e-9


@settings(max_examples=200, deadline=None)
@given(st.integers(min_value=1, max_value=10 ** 6),
       st.integers(min_value=1, max_value=64))
def test_shard_policy_partitions_exactly(total, nshards):
    sizes = [shard_size(total, nshards, s) for s in range(nshards)]
    offs = [shard_offset(total, nshards, s) for s in range(nshards)]
    assert sum(sizes) == total
    # offsets are the prefix sums (contiguous, disjoint, ordered)
    acc = 0
    for o, sz in zip(offs, sizes):
        assert o == acc
        acc += sz
    # remainder-to-shard-0 policy
    assert sizes[0] == max(sizes)


@settings(max_examples=200, deadline=None)
@given(st.integers(min_value=1, max_value=200),
       st.integers(min_value=1, max_value=16),
       st.booleans(), st.booleans())
def test_slice_files_partitions(n, reps, unbalanced, drop_last):
    files = list(range(n))
    if n < reps and not (drop_last and False):
        # under-filled: must raise (unless drop_last empties... it
        # cannot — drop_last only trims the remainder)
        if reps > 1 and (n - (n % reps if drop_last else 0)) < reps:
            try:
                slice_files(files, reps, 0, unbalanced, drop_last)
            except ValueError:
                return
    slices = [slice_files(files, reps, r, unbalanced, drop_last)
              for r in range(reps)]
    flat = [f for s in slices for f in s]
    if drop_last and reps > 1:
        assert flat == files[:n - n % reps]
        assert len(set(len(s) for s in slices)) == 1
    else:
        assert flat == files
        assert max(len(s) for s in slices) - min(len(s)
                                                 for s in slices) <= 1


@settings(max_examples=60, deadline=None)
@given(st.lists(st.integers(min_value=1, max_value=2000), min_size=1,
                max_size=12),
       st.integers(min_value=1, max_value=8),
       st.integers(min_value=4096, max_value=10 ** 6))
def test_bucket_layout_random(param_sizes, world, bucket_bytes):
    """GradReducer bucket invariants over random param sets: buckets
    tile the arena exactly; every param's extent is covered by its
    gating buckets; ZeRO shard edges force boundaries."""
    import torch.nn as nn

    from easyparallellibrary_amd.comm.pool import CommunicationPool
    from easyparallellibrary_amd.parallel.dp import (FlatParamGroup,
                                                     GradReducer, _aligned)
    params = [nn.Parameter(torch.zeros(s)) for s in param_sizes]
    fg = FlatParamGroup(params, device="cpu",
                        pad_to_multiple=world if world > 1 else 1)

    class _FakeComm:
        size = world
        rank = 0

        def all_reduce(self, *a, **k):
            pass

        def reduce(self, *a, **k):
            pass

    class _FakePool:
        comms = [_FakeComm()]
        size = 1

    red = GradReducer(fg, _FakePool(), bucket_bytes, overlap=False,
                      shard_owners=(world > 1))
    pos = 0
    for (start, end, ps, owner) in red.buckets:
        assert start == pos and end > start
        pos = end
        if world > 1:
            shard = fg.total // world
            assert end <= ((start // shard) + 1) * shard  # no shard straddle
            assert owner == min(start // shard, world - 1)
    assert pos == fg.total
    for p, off in zip(fg.ordered, fg.offsets):
        lo, hi = off, off + _aligned(p.numel())
        gated = [b for b in red.buckets if any(q is p for q in b[2])]
        assert gated
        assert min(b[0] for b in gated) <= lo
        assert max(b[1] for b in gated) >= hi
    red.remove_hooks()


def test_find_repeated_blocks_prefers_dominant():
    assert find_repeated_blocks(list("aabbbbcc"), key=lambda x: x) == "b"
    assert find_repeated_blocks(list("ab"), key=lambda x: x) is None
    assert find_repeated_blocks(list("abc") * 3, key=lambda x: x) == "a"
