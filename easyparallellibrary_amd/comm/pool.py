"""Round-robin pool of communicators.

Capability parity: /root/reference/epl/communicators/communication_pool.py
(:84-105 — N communicators, buckets round-robined, per-communicator
serialization).  On MI355X each pool member owns its own HIP stream and RCCL
comm, so K pool members = K concurrent rings over distinct xGMI links;
serialization within a member comes for free from its stream order.
Default pool size: config.communication.num_communicators (4; reference 2).
"""

from easyparallellibrary_amd.comm.backend import create_communicator


class CommunicationPool:
    def __init__(self, base_name, ranks, num_communicators=2):
        self.ranks = list(ranks)
        self.comms = [
            create_communicator("{}_p{}".format(base_name, i), self.ranks)
            for i in range(max(1, num_communicators))
        ]
        self._next = 0

    @property
    def size(self):
        return len(self.comms)

    def next_comm(self):
        comm = self.comms[self._next]
        self._next = (self._next + 1) % len(self.comms)
        return comm

    def batch_all_reduce(self, buckets, op="sum", async_op=True):
        """All-reduce a list of contiguous buckets, round-robined over the
        pool (reference: collective_communicator.py:93-123).

        Mapping is positional (bucket i -> comm i mod K), NOT the stateful
        next_comm() cursor: every rank must issue bucket i on the same
        communicator, and cursor state could desync if call counts ever
        differed across ranks (e.g. a rank-gated extra collective)."""
        for i, b in enumerate(buckets):
            self.comms[i % len(self.comms)].all_reduce(
                b, op=op, async_op=async_op)

    def join(self):
        for c in self.comms:
            c.join()

    def synchronize(self):
        for c in self.comms:
            c.synchronize()
