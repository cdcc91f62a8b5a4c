"""Collective facade on gloo (reference: tests/communicator_test.py
numeric checks) + autograd collective gradients
(reference: epl/communicators/nccl_ops.py:37-124 registrations)."""

import torch

from tests.utils import run_multiprocess


def _collectives_worker(rank, world):
    import torch.distributed as dist
    from easyparallellibrary_amd.env import Env
    Env.get().get_or_create_process_group()
    from easyparallellibrary_amd.comm.backend import create_communicator
    comm = create_communicator("t_c", list(range(world)))

    out = {}
    t = torch.full((4,), float(rank + 1))
    comm.all_reduce(t)
    out["allreduce"] = t.clone()

    t = torch.full((2,), float(rank))
    g = torch.empty(2 * world)
    comm.all_gather(g, t)
    out["allgather"] = g.clone()

    t = torch.arange(4.0) + rank * 4
    o = torch.empty(4 // world * 2)
    # reduce_scatter over [world * k] input
    inp = torch.full((world * 2,), float(rank + 1))
    o = torch.empty(2)
    comm.reduce_scatter(o, inp)
    out["reduce_scatter"] = o.clone()

    t = torch.full((3,), float(rank + 5))
    comm.broadcast(t, root=1)
    out["broadcast"] = t.clone()

    inp = torch.arange(float(world * 2)) + rank * 100
    o = torch.empty(world * 2)
    comm.all_to_all_single(o, inp)
    out["a2a"] = o.clone()

    # all_to_all_v: rank sends (r+1) elements to each rank r
    in_counts = [rank + 1] * world  # wrong: counts per dest
    # simpler: each rank sends 1 element to every rank
    inp = torch.full((world,), float(rank))
    o = torch.empty(world)
    comm.all_to_all_v(o, inp, [1] * world, [1] * world)
    out["a2av"] = o.clone()

    # all_gather_v: rank r contributes r+1 elements of value r
    mine = torch.full((rank + 1,), float(rank))
    outs = [torch.empty(r + 1) for r in range(world)]
    comm.all_gather_v(outs, mine)
    out["agv"] = torch.cat(outs)
    return out


def test_collectives_numeric():
    r = run_multiprocess(_collectives_worker, world=2)
    assert torch.equal(r[0]["allreduce"], torch.full((4,), 3.0))
    assert torch.equal(r[1]["allreduce"], torch.full((4,), 3.0))
    assert torch.equal(r[0]["allgather"],
                       torch.tensor([0., 0., 1., 1.]))
    assert torch.equal(r[0]["reduce_scatter"], torch.full((2,), 3.0))
    assert torch.equal(r[1]["reduce_scatter"], torch.full((2,), 3.0))
    assert torch.equal(r[0]["broadcast"], torch.full((3,), 6.0))
    # a2a: rank0 out = [own first half, rank1 first half]
    assert torch.equal(r[0]["a2a"], torch.tensor([0., 1., 100., 101.]))
    assert torch.equal(r[1]["a2a"], torch.tensor([2., 3., 102., 103.]))
    assert torch.equal(r[0]["a2av"], torch.tensor([0., 1.]))
    assert torch.equal(r[1]["a2av"], torch.tensor([0., 1.]))
    # all_gather_v: [rank0's 1 elem, rank1's 2 elems] on every rank
    assert torch.equal(r[0]["agv"], torch.tensor([0., 1., 1.]))
    assert torch.equal(r[1]["agv"], torch.tensor([0., 1., 1.]))


def _autograd_worker(rank, world):
    from easyparallellibrary_amd.env import Env
    Env.get().get_or_create_process_group()
    from easyparallellibrary_amd.comm import functional
    from easyparallellibrary_amd.comm.backend import create_communicator
    comm = create_communicator("t_ag", list(range(world)))
    out = {}

    x = torch.full((2, 3), float(rank + 1), requires_grad=True)
    y = functional.all_gather(x, comm)       # [2*world, 3]
    (y.sum() * (rank + 1)).backward()        # dL/dy = rank+1 everywhere
    # backward = reduce_scatter(sum of per-rank grads) = sum over ranks
    out["ag_grad"] = x.grad.clone()

    x = torch.full((world * 2, 3), float(rank + 1), requires_grad=True)
    y = functional.reduce_scatter(x, comm)
    y.sum().backward()                       # backward = all_gather of ones
    out["rs_grad"] = x.grad.clone()

    x = torch.full((4,), float(rank), requires_grad=True)
    y = functional.all_reduce(x, comm)
    (y * 2).sum().backward()
    out["ar_grad"] = x.grad.clone()

    x = torch.full((world, 2), float(rank), requires_grad=True)
    y = functional.all_to_all(x, comm)
    (y.sum(dim=1) * torch.arange(1.0, world + 1)).sum().backward()
    out["a2a_grad"] = x.grad.clone()
    return out


def test_autograd_collectives():
    r = run_multiprocess(_autograd_worker, world=2)
    # all_gather backward: grad = sum over ranks of their dL/dy slice for me
    assert torch.equal(r[0]["ag_grad"], torch.full((2, 3), 3.0))
    assert torch.equal(r[1]["ag_grad"], torch.full((2, 3), 3.0))
    # reduce_scatter backward: allgather of ones -> all ones
    assert torch.equal(r[0]["rs_grad"], torch.ones(4, 3))
    # all_reduce backward: allreduce of grads (2 everywhere -> 4)
    assert torch.equal(r[0]["ar_grad"], torch.full((4,), 4.0))
    # a2a backward: grad routed back: my row r gets weight (my_row sent to
    # rank r, weighted there by (r+1) at position of my rank)
    assert torch.equal(r[0]["a2a_grad"],
                       torch.tensor([[1., 1.], [1., 1.]]))
    assert torch.equal(r[1]["a2a_grad"],
                       torch.tensor([[2., 2.], [2., 2.]]))
