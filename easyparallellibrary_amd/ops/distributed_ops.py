"""Sharded argmax / equal (accuracy metrics over vocab-sharded logits).

Capability parity: /root/reference/epl/ops/distributed_ops.py
(DistributedArgmax two-level max :58-95, distributed equal :125-148).
"""

import torch

from easyparallellibrary_amd.comm import functional


def distributed_argmax(logits, comm, vocab_begin=0):
    """Global argmax over the sharded last dim: local (max, idx) ->
    allgather -> global pick."""
    local_max, local_idx = logits.max(dim=-1)
    local_idx = local_idx + vocab_begin
    if comm is None or comm.size == 1:
        return local_idx
    flat_max = local_max.reshape(-1).contiguous().float()
    flat_idx = local_idx.reshape(-1).contiguous().to(torch.int64)
    gmax = functional.all_gather(flat_max.unsqueeze(0).contiguous(), comm)
    gidx = functional.all_gather(flat_idx.unsqueeze(0).contiguous(), comm)
    winner = gmax.argmax(dim=0)
    out = gidx.gather(0, winner.unsqueeze(0)).squeeze(0)
    return out.reshape(local_idx.shape)


def distributed_equal(pred, labels, comm=None, vocab_begin=0):
    """Accuracy helper: pred from distributed_argmax vs labels."""
    return (pred == labels)
