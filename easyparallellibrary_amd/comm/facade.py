"""Per-use-case collective facade.

Capability parity: /root/reference/epl/communicators/collective_communicator.py
(batch_allreduce :93-123, broadcast :125-141, allgather :143-149, alltoall
:151-157, reduce :159-177, bucket-size estimation :183-204).
"""

import torch

from easyparallellibrary_amd.comm import functional
from easyparallellibrary_amd.comm.backend import create_communicator
from easyparallellibrary_amd.comm.pool import CommunicationPool


def estimate_split_num_for_comm(total_bytes, bucket_bytes, max_splits):
    """Bucket count for a payload (reference :183-204: 32 MB target).  On
    MI355X the pool overlaps buckets on distinct xGMI rings, so we aim for
    at least pool-size buckets when the payload is large."""
    if total_bytes <= 0:
        return 1
    n = (total_bytes + bucket_bytes - 1) // bucket_bytes
    return max(1, min(int(n), int(max_splits)))


class CollectiveCommunicator:
    """One logical communication context over a rank group: a pool for
    bulk gradient traffic plus a dedicated comm for inline ops."""

    def __init__(self, name, ranks, num_communicators=1):
        self.name = name
        self.ranks = list(ranks)
        self.pool = CommunicationPool(name, ranks, num_communicators)
        self.inline = create_communicator("{}_inline".format(name), ranks)

    @property
    def size(self):
        return self.inline.size

    @property
    def rank(self):
        return self.inline.rank

    # bulk (non-autograd) ------------------------------------------------------
    def batch_allreduce(self, buckets, op="sum", async_op=True):
        self.pool.batch_all_reduce(buckets, op=op, async_op=async_op)

    def broadcast_weights(self, tensors, root=0):
        """Serial bucketed broadcast for initial weight sync
        (reference: hooks.py:330-357)."""
        for t in tensors:
            self.inline.broadcast(t, root=root)

    def join(self):
        self.pool.join()

    def synchronize(self):
        self.pool.synchronize()
        self.inline.synchronize()

    # inline autograd ops ------------------------------------------------------
    def all_reduce(self, t, op="sum"):
        return functional.all_reduce(t, self.inline, op)

    def all_gather(self, t):
        return functional.all_gather(t, self.inline)

    def reduce_scatter(self, t, op="sum"):
        return functional.reduce_scatter(t, self.inline, op)

    def all_to_all(self, t):
        return functional.all_to_all(t, self.inline)

    def all_gather_v(self, t, sizes):
        """Varying-size all-gather (reference AllGatherv): ``sizes[r]`` is
        rank r's dim-0 length; returns the list of per-rank tensors."""
        outs = [torch.empty((int(n),) + tuple(t.shape[1:]), dtype=t.dtype,
                            device=t.device) for n in sizes]
        return self.inline.all_gather_v(outs, t.contiguous())

    def broadcast(self, t, root=0):
        return functional.broadcast(t, self.inline, root)
