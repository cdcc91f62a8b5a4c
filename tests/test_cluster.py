"""Cluster / layout (reference: tests/cluster_test.py semantics)."""

import pytest

from easyparallellibrary_amd.cluster import Layout, SpecificLayout


def test_pure_dp_layout():
    lo = Layout(8, [1])
    vds = lo.slices()
    assert lo.num_replicas == 8
    assert vds[0].ranks_per_replica == [[r] for r in range(8)]


def test_pp2_dp4_layout():
    lo = Layout(8, [1, 1])
    assert lo.num_replicas == 4
    vds = lo.slices()
    # replica-contiguous: replica r = ranks [2r, 2r+1]
    assert vds[0].ranks_per_replica == [[0], [2], [4], [6]]
    assert vds[1].ranks_per_replica == [[1], [3], [5], [7]]


def test_split8_layout():
    lo = Layout(8, [8])
    assert lo.num_replicas == 1
    assert lo.slices()[0].ranks_per_replica == [list(range(8))]


def test_indivisible_raises():
    with pytest.raises(ValueError):
        Layout(8, [1, 1, 1])


def test_specific_layout():
    sl = SpecificLayout([[[0], [1]], [[2], [3]]])
    vds = sl.slices()
    assert vds[0].local_ranks(1) == [1]
    assert vds[1].replica_of_rank(3) == 1


def test_layout_tiling_randomized():
    """Layout.slices() covers every rank exactly once per replica set."""
    import random
    rng = random.Random(5)
    for _ in range(100):
        counts = [rng.randint(1, 4) for _ in range(rng.randint(1, 5))]
        per = sum(counts)
        replicas = rng.randint(1, 4)
        world = per * replicas
        lay = Layout(world, counts)
        assert lay.num_replicas == replicas
        seen = set()
        for vd in lay.slices():
            for rep in range(replicas):
                for r in vd.local_ranks(rep):
                    assert r not in seen
                    seen.add(r)
        assert seen == set(range(world))
