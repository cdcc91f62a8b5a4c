"""Cluster and device model.

Capability parity: /root/reference/epl/cluster.py (Cluster :295-370,
VirtualDevice :36-100, AllLayout :108, AutoLayout :146, SpecificLayout :162,
generate_device_slices :133 honouring device_place_prefer_intra_node).

MI355X-native redesign: instead of TF device strings and a TF_CONFIG grpc
mesh, the process model is one rank per GPU under torch.distributed (RCCL
over xGMI).  A ``VirtualDevice`` is a slice of global ranks assigned to one
taskgraph (pipeline stage / split scope); ``Layout`` decides how the world's
ranks are carved into (replica, stage) coordinates.  Replicas own contiguous
rank ranges so that a replica's pipeline stages sit on one node and p2p
activations ride a single xGMI hop.
"""

import os

import torch


class VirtualDevice:
    """Devices assigned to one taskgraph: for every replica, the list of
    global ranks that execute this taskgraph (reference: epl/cluster.py:36-100).
    """

    def __init__(self, taskgraph_index, ranks_per_replica):
        self.taskgraph_index = taskgraph_index
        # list over replicas; each entry is a list of global ranks
        self.ranks_per_replica = [list(r) for r in ranks_per_replica]

    @property
    def num_replicas(self):
        return len(self.ranks_per_replica)

    @property
    def all_ranks(self):
        return [r for rep in self.ranks_per_replica for r in rep]

    def local_ranks(self, replica_idx):
        return self.ranks_per_replica[replica_idx]

    def replica_of_rank(self, rank):
        for i, rep in enumerate(self.ranks_per_replica):
            if rank in rep:
                return i
        return None

    def __repr__(self):
        return "VirtualDevice(tg={}, replicas={})".format(
            self.taskgraph_index, self.ranks_per_replica)


class Layout:
    """Carve ``world_size`` ranks into per-taskgraph VirtualDevices.

    ``device_counts`` is the per-taskgraph ``device_count`` from the user's
    annotations (1 for a plain replicate stage, N for an N-way split scope).
    A *replica* needs ``sum(device_counts)`` ranks; the number of replicas is
    ``world_size // per_replica`` (reference AutoLayout, epl/cluster.py:146-159).

    Deliberately NOT ported: the reference's AwareRowLayout host
    reordering (epl/cluster.py:169-241) permutes worker rows so
    cross-HOST pipeline edges minimize inter-node traffic.  This
    framework targets one 8-GPU MI355X node where every pair of GPUs is
    one xGMI hop, so host-aware reordering has nothing to optimize;
    multi-node jobs should order ranks host-contiguously at launch
    (the launcher already assigns ranks per worker in order), which
    yields the same placement AwareRowLayout computes for the
    homogeneous case.

    Ranks are laid out replica-major: replica ``r`` owns the contiguous range
    ``[r*per_replica, (r+1)*per_replica)``; inside a replica, taskgraphs are
    packed in order.  This keeps each replica's pipeline chain on adjacent
    ranks (intra-node xGMI) and makes every DP group a constant-stride rank
    set, which RCCL turns into parallel rings over distinct xGMI links.
    """

    def __init__(self, world_size, device_counts):
        self.world_size = world_size
        self.device_counts = list(device_counts)
        per_replica = sum(self.device_counts)
        if per_replica <= 0:
            raise ValueError("need at least one taskgraph device")
        if world_size % per_replica != 0:
            raise ValueError(
                "world_size {} not divisible by devices-per-replica {} "
                "(taskgraph device_counts={})".format(
                    world_size, per_replica, self.device_counts))
        self.per_replica = per_replica
        self.num_replicas = world_size // per_replica

    def slices(self):
        out = []
        offset = 0
        for tg, count in enumerate(self.device_counts):
            reps = []
            for r in range(self.num_replicas):
                base = r * self.per_replica + offset
                reps.append(list(range(base, base + count)))
            out.append(VirtualDevice(tg, reps))
            offset += count
        return out


class SpecificLayout:
    """Explicit user-provided rank map: list over taskgraphs of list over
    replicas of rank lists (reference: epl/cluster.py:162)."""

    def __init__(self, rank_map):
        self.rank_map = rank_map

    def slices(self):
        return [VirtualDevice(i, reps) for i, reps in enumerate(self.rank_map)]


class Cluster:
    """The distributed world: rank/world_size/local device, plus the
    device-slicing entry point (reference: epl/cluster.py:295-370).

    Works in three modes:
      * torch.distributed already initialised -> adopt its rank/world.
      * env RANK/WORLD_SIZE set (torchrun / epl-launch) -> use env.
      * neither -> single-process world of size 1.
    """

    def __init__(self, worker_hosts=None):
        del worker_hosts  # single-node xGMI world; kept for API parity
        import torch.distributed as dist
        if dist.is_available() and dist.is_initialized():
            self.world_size = dist.get_world_size()
            self.rank = dist.get_rank()
        else:
            self.world_size = int(os.environ.get("WORLD_SIZE", "1"))
            self.rank = int(os.environ.get("RANK", "0"))
        self.local_rank = int(os.environ.get("LOCAL_RANK", self.rank))
        self.gpu_available = torch.cuda.is_available()
        if self.gpu_available:
            self.num_local_gpus = torch.cuda.device_count()
            self.device = torch.device("cuda", self.local_rank % self.num_local_gpus)
        else:
            self.num_local_gpus = 0
            self.device = torch.device("cpu")
        self.virtual_devices = []

    @property
    def total_gpu_num(self):
        return self.world_size if self.gpu_available else 0

    def set_virtual_devices(self, device_counts, rank_map=None):
        """Assign VirtualDevices for the given per-taskgraph device counts."""
        if rank_map is not None:
            layout = SpecificLayout(rank_map)
            self.virtual_devices = layout.slices()
        else:
            layout = Layout(self.world_size, device_counts)
            self.virtual_devices = layout.slices()
        return self.virtual_devices

    def __repr__(self):
        return "Cluster(rank={}/{}, device={})".format(
            self.rank, self.world_size, self.device)
