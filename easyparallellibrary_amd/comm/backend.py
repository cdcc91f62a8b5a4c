"""Communicator backends.

Capability parity: /root/reference/epl/communicators/base.py (abstract
Communicator :149-259, bootstrap id broadcast :43-73) and nccl.py
(NcclCommunicator :78-231).

MI355X redesign: the GPU backend is our own C++ RCCL core
(csrc/comm/rccl_comm.hip) — each communicator owns a ncclComm_t plus a
dedicated HIP stream; the unique-id bootstrap rides the torch.distributed
TCP store instead of TF collective_ops.  On CPU (unit tests, plumbing
config 1 of BASELINE.json) a gloo-backed communicator implements the same
interface; a single-rank communicator short-circuits everything locally.
"""

import hashlib

import torch
import torch.distributed as dist

_REGISTRY = {}


def _ext():
    from easyparallellibrary_amd import _C
    return _C


class Communicator:
    """Abstract collective interface (reference: base.py:149-259)."""

    def __init__(self, name, ranks):
        self.name = name
        self.ranks = list(ranks)
        self.size = len(self.ranks)
        self.global_rank = dist.get_rank() if dist.is_initialized() else 0
        self.rank = (self.ranks.index(self.global_rank)
                     if self.global_rank in self.ranks else -1)

    # collective API -----------------------------------------------------------
    def all_reduce(self, t, op="sum", async_op=False):
        raise NotImplementedError

    def broadcast(self, t, root=0, async_op=False):
        raise NotImplementedError

    def reduce(self, t, root=0, op="sum", async_op=False):
        raise NotImplementedError

    def all_gather(self, out, inp, async_op=False):
        raise NotImplementedError

    def reduce_scatter(self, out, inp, op="sum", async_op=False):
        raise NotImplementedError

    def all_to_all_single(self, out, inp, async_op=False):
        raise NotImplementedError

    def all_to_all_v(self, out, inp, out_counts, in_counts, async_op=False):
        raise NotImplementedError

    def all_gather_v(self, outs, inp, async_op=False):
        """Varying-size all-gather: rank r's ``inp`` lands in ``outs[r]``
        on every rank (reference AllGatherv, tensorflow_nccl.h:169-183:
        a grouped per-rank broadcast — sizes may differ per rank)."""
        raise NotImplementedError

    def send(self, t, peer):
        raise NotImplementedError

    def recv(self, t, peer):
        raise NotImplementedError

    def batch_p2p(self, ops):
        """ops: list of (is_send, tensor, peer_group_rank)."""
        raise NotImplementedError

    def join(self):
        """Fence: current stream waits for this communicator's stream."""

    def synchronize(self):
        """Host-sync this communicator's stream."""


class LocalCommunicator(Communicator):
    """size==1: collectives are identities/copies."""

    def __init__(self, name, ranks):
        super().__init__(name, ranks)
        self.rank = 0
        self.size = 1

    def all_reduce(self, t, op="sum", async_op=False):
        return t

    def broadcast(self, t, root=0, async_op=False):
        return t

    def reduce(self, t, root=0, op="sum", async_op=False):
        return t

    def all_gather(self, out, inp, async_op=False):
        out.copy_(inp.reshape(out.shape))
        return out

    def reduce_scatter(self, out, inp, op="sum", async_op=False):
        out.copy_(inp.reshape(out.shape))
        return out

    def all_to_all_single(self, out, inp, async_op=False):
        out.copy_(inp)
        return out

    def all_to_all_v(self, out, inp, out_counts, in_counts, async_op=False):
        out.copy_(inp.reshape(out.shape))
        return out

    def all_gather_v(self, outs, inp, async_op=False):
        outs[0].copy_(inp.reshape(outs[0].shape))
        return outs


class RcclCommunicator(Communicator):
    """GPU communicator backed by the native RCCL core.

    Bootstrap (reference: base.py:43-73 uses TF collective_ops): the group
    root creates a ncclUniqueId and publishes it on the torch.distributed
    TCP store under a key derived from (name, ranks); members fetch it and
    call ncclCommInitRank.
    """

    def __init__(self, name, ranks):
        super().__init__(name, ranks)
        assert self.rank >= 0, "rank {} not in group {}".format(
            self.global_rank, ranks)
        ext = _ext()
        tag = hashlib.md5(
            ("{}:{}".format(name, ",".join(map(str, self.ranks))))
            .encode()).hexdigest()[:16]
        if self.size == 1 or not dist.is_initialized():
            uid = ext.comm_unique_id()
        else:
            store = dist.distributed_c10d._get_default_store()
            key = "epl/comm/{}".format(tag)
            if self.rank == 0:
                uid = ext.comm_unique_id()
                store.set(key, uid.hex())
            else:
                uid = bytes.fromhex(store.get(key).decode())
        ext.comm_init(name, uid, self.rank, self.size)

    def _peer(self, peer):
        return int(peer)

    def all_reduce(self, t, op="sum", async_op=False):
        _ext().all_reduce(self.name, t, op, async_op)
        return t

    def broadcast(self, t, root=0, async_op=False):
        _ext().broadcast(self.name, t, root, async_op)
        return t

    def reduce(self, t, root=0, op="sum", async_op=False):
        _ext().reduce(self.name, t, root, op, async_op)
        return t

    def all_gather(self, out, inp, async_op=False):
        _ext().all_gather(self.name, out, inp, async_op)
        return out

    def reduce_scatter(self, out, inp, op="sum", async_op=False):
        _ext().reduce_scatter(self.name, out, inp, op, async_op)
        return out

    def all_to_all_single(self, out, inp, async_op=False):
        _ext().all_to_all_single(self.name, out, inp, async_op)
        return out

    def all_to_all_v(self, out, inp, out_counts, in_counts, async_op=False):
        _ext().all_to_all_v(self.name, out, inp, list(out_counts),
                            list(in_counts), async_op)
        return out

    def all_gather_v(self, outs, inp, async_op=False):
        _ext().all_gather_v(self.name, list(outs), inp, async_op)
        return outs

    def send(self, t, peer):
        _ext().send(self.name, t, self._peer(peer), False)

    def recv(self, t, peer):
        _ext().recv(self.name, t, self._peer(peer), False)

    def batch_p2p(self, ops):
        _ext().batch_p2p(
            self.name, [(s, t, int(p)) for (s, t, p) in ops], False)

    def join(self):
        _ext().comm_join(self.name)

    def synchronize(self):
        _ext().comm_synchronize(self.name)

    def destroy(self):
        _ext().comm_destroy(self.name)


class GlooCommunicator(Communicator):
    """CPU fallback over torch.distributed gloo groups.  Collectives the
    gloo backend lacks (reduce_scatter, all_to_all) are emulated so the
    full facade is testable without a GPU."""

    def __init__(self, name, ranks):
        super().__init__(name, ranks)
        if dist.is_initialized():
            self.group = (dist.new_group(self.ranks)
                          if self.size < dist.get_world_size()
                          else dist.group.WORLD)
        else:
            self.group = None

    def _gop(self, op):
        return {
            "sum": dist.ReduceOp.SUM,
            "prod": dist.ReduceOp.PRODUCT,
            "max": dist.ReduceOp.MAX,
            "min": dist.ReduceOp.MIN,
        }[op]

    def all_reduce(self, t, op="sum", async_op=False):
        if op == "avg":
            dist.all_reduce(t, dist.ReduceOp.SUM, group=self.group)
            t.div_(self.size)
        else:
            dist.all_reduce(t, self._gop(op), group=self.group)
        return t

    def broadcast(self, t, root=0, async_op=False):
        dist.broadcast(t, self.ranks[root], group=self.group)
        return t

    def reduce(self, t, root=0, op="sum", async_op=False):
        if op == "avg":
            dist.reduce(t, self.ranks[root], dist.ReduceOp.SUM,
                        group=self.group)
            if self.rank == root:
                t.div_(self.size)
        else:
            dist.reduce(t, self.ranks[root], self._gop(op), group=self.group)
        return t

    def all_gather(self, out, inp, async_op=False):
        chunks = list(out.reshape(self.size, -1).unbind(0))
        dist.all_gather(chunks, inp.reshape(-1), group=self.group)
        return out

    def reduce_scatter(self, out, inp, op="sum", async_op=False):
        tmp = inp.clone()
        self.all_reduce(tmp, op)
        out.copy_(tmp.reshape(self.size, -1)[self.rank].reshape(out.shape))
        return out

    def all_to_all_single(self, out, inp, async_op=False):
        # pairwise p2p (O(W) traffic per rank), not allgather (O(W^2)):
        # keeps CPU plumbing tests representative of the RCCL a2a cost
        chunk = inp.numel() // self.size
        counts = [chunk] * self.size
        return self.all_to_all_v(out, inp, counts, counts, async_op)

    def all_to_all_v(self, out, inp, out_counts, in_counts, async_op=False):
        reqs = []
        in_off = 0
        flat_in = inp.reshape(-1)
        for r in range(self.size):
            n = int(in_counts[r])
            if n > 0:
                if r == self.rank:
                    pass
                else:
                    reqs.append(dist.isend(
                        flat_in[in_off:in_off + n].contiguous(),
                        self.ranks[r], group=self.group))
            in_off += n
        out_off = 0
        flat_out = out.reshape(-1)
        my_in_off = sum(int(c) for c in in_counts[:self.rank])
        for r in range(self.size):
            n = int(out_counts[r])
            if n > 0:
                if r == self.rank:
                    flat_out[out_off:out_off + n].copy_(
                        flat_in[my_in_off:my_in_off + n])
                else:
                    buf = torch.empty(n, dtype=inp.dtype, device=inp.device)
                    dist.recv(buf, self.ranks[r], group=self.group)
                    flat_out[out_off:out_off + n].copy_(buf)
            out_off += n
        for rq in reqs:
            rq.wait()
        return out

    def all_gather_v(self, outs, inp, async_op=False):
        for r in range(self.size):
            if r == self.rank:
                # copy_ on the tensor itself: reshape(-1) on a
                # non-contiguous out would return a copy and drop the
                # write (advisor finding r1)
                outs[r].copy_(inp.reshape(outs[r].shape))
            dist.broadcast(outs[r], self.ranks[r], group=self.group)
        return outs

    def send(self, t, peer):
        dist.send(t, self.ranks[peer], group=self.group)

    def recv(self, t, peer):
        dist.recv(t, self.ranks[peer], group=self.group)

    def batch_p2p(self, ops):
        reqs = []
        for is_send, t, peer in ops:
            if is_send:
                reqs.append(dist.isend(t, self.ranks[int(peer)],
                                       group=self.group))
            else:
                reqs.append(dist.irecv(t, self.ranks[int(peer)],
                                       group=self.group))
        for rq in reqs:
            rq.wait()


def create_communicator(name, ranks, device=None):
    """Factory (reference: parallel/ops.py:421-451).  Caches by name."""
    if name in _REGISTRY:
        return _REGISTRY[name]
    ranks = list(ranks)
    if len(ranks) <= 1:
        comm = LocalCommunicator(name, ranks)
    elif torch.cuda.is_available():
        comm = RcclCommunicator(name, ranks)
    else:
        comm = GlooCommunicator(name, ranks)
    _REGISTRY[name] = comm
    return comm


def get_communicator(name):
    return _REGISTRY.get(name)


def destroy_all_communicators():
    for name, comm in list(_REGISTRY.items()):
        if isinstance(comm, RcclCommunicator):
            comm.destroy()
        del _REGISTRY[name]


def destroy_namespace(prefix):
    """Destroy every communicator whose name starts with ``prefix`` —
    Engine.close() uses this to reclaim RCCL comms + HIP streams (the
    registry otherwise grows for the life of the process)."""
    n = 0
    for name, comm in list(_REGISTRY.items()):
        if name.startswith(prefix):
            if isinstance(comm, RcclCommunicator):
                comm.destroy()
            del _REGISTRY[name]
            n += 1
    return n
