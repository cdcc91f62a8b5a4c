"""Replica<->Split bridging.

Capability parity: /root/reference/epl/ops/bridging_layer.py:46-58
(Replica2Split: allgather a replica-local tensor to the full batch before a
split scope; backward reduce-scatters automatically via the autograd
collective).
"""

from easyparallellibrary_amd.comm import functional


def replica_to_split(x, comm):
    """Gather the per-rank local batch into the full batch (dim 0) so a
    split (sharded) scope sees every replica's samples."""
    return functional.all_gather(x, comm)


def split_to_replica(x, comm):
    """Take this replica's slice of a full-batch tensor produced by a split
    scope (inverse bridge).  Backward is the plain slice adjoint (zeros
    outside this replica's rows); the other replicas' contributions to
    the split scope's parameter gradients arrive through the scope's own
    cross-rank gradient reduction, mirroring the reference's
    Split2Replica."""
    if comm.size == 1:
        return x
    n = x.shape[0] // comm.size
    return x[comm.rank * n:(comm.rank + 1) * n]


def replica_to_replica(x, comm):
    """Bridge between two replicate scopes on the same devices: identity
    (reference stub, bridging_layer.py Replica2Replica)."""
    return x


def split_to_split(x, src_comm, dst_comm):
    """Bridge between two split scopes; equal degrees pass through
    (reference stub, bridging_layer.py Split2Split)."""
    if (src_comm is None) != (dst_comm is None):
        raise ValueError("split_to_split needs both communicators")
    if src_comm is not None and src_comm.size != dst_comm.size:
        raise NotImplementedError(
            "re-sharding between different split degrees")
    return x
