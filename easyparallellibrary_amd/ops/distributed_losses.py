"""Sharded-vocab softmax cross-entropy.

Capability parity: /root/reference/epl/ops/distributed_losses.py
(DistributedSoftmaxCrossEntropy :58-109: allgather local max -> global max,
exp, allreduce of normalizers, label-range mask, final loss allreduce).

MI355X redesign: the per-shard row statistics (max, sum-exp, target logit)
are ONE hand-written CDNA4 kernel pass over the local logits
(csrc/kernels/kernels.hip: ce_rowstats); only three tiny [rows] fp32
vectors cross xGMI (max: allreduce-max, rescaled sumexp + target logit:
allreduce-sum).  The backward writes dlogits in a single kernel pass.
CPU fallback implements identical math for the numerics tests.
"""

import torch

from easyparallellibrary_amd.ops.dispatch import native_ext, use_native


class _VocabParallelCE(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, targets, comm, vocab_begin, ignore_index):
        rows = logits.shape[0]
        cols = logits.shape[1]
        dev = logits.device
        if use_native(logits):
            ext = native_ext()
            lmax = torch.empty(rows, dtype=torch.float32, device=dev)
            lsum = torch.empty(rows, dtype=torch.float32, device=dev)
            tlogit = torch.empty(rows, dtype=torch.float32, device=dev)
            ext.ce_rowstats(logits, targets, lmax, lsum, tlogit, vocab_begin,
                            ignore_index)
        else:
            lf = logits.float()
            lmax = lf.max(dim=1).values
            lsum = torch.exp(lf - lmax[:, None]).sum(dim=1)
            in_range = (targets != ignore_index) & (targets >= vocab_begin) \
                & (targets < vocab_begin + cols)
            safe = (targets - vocab_begin).clamp(0, cols - 1)
            tlogit = torch.where(
                in_range, lf.gather(1, safe[:, None]).squeeze(1),
                torch.zeros(rows, device=dev))
        if comm is not None and comm.size > 1:
            gmax = lmax.clone()
            comm.all_reduce(gmax, op="max")
            gsum = lsum * torch.exp(lmax - gmax)
            comm.all_reduce(gsum, op="sum")
            comm.all_reduce(tlogit, op="sum")
        else:
            gmax, gsum = lmax, lsum
        valid = (targets != ignore_index)
        losses = torch.where(
            valid, torch.log(gsum) + gmax - tlogit,
            torch.zeros_like(gmax))
        ctx.save_for_backward(logits, targets, gmax, gsum)
        ctx.comm = comm
        ctx.vocab_begin = vocab_begin
        ctx.ignore_index = ignore_index
        return losses

    @staticmethod
    def backward(ctx, dloss):
        logits, targets, gmax, gsum = ctx.saved_tensors
        dloss = dloss.contiguous().float()
        if use_native(logits):
            dlogits = torch.empty_like(logits)
            native_ext().ce_backward(dlogits, logits, targets, gmax, gsum,
                                     dloss, ctx.vocab_begin,
                                     ctx.ignore_index, 1.0)
        else:
            lf = logits.float()
            p = torch.exp(lf - gmax[:, None]) / gsum[:, None]
            cols = logits.shape[1]
            in_range = (targets != ctx.ignore_index) & \
                (targets >= ctx.vocab_begin) & \
                (targets < ctx.vocab_begin + cols)
            safe = (targets - ctx.vocab_begin).clamp(0, cols - 1)
            onehot = torch.zeros_like(p)
            onehot[in_range, safe[in_range]] = 1.0
            mask = (targets != ctx.ignore_index).float()[:, None]
            dlogits = ((p - onehot) * dloss[:, None] * mask).to(logits.dtype)
        return dlogits, None, None, None, None


def vocab_parallel_cross_entropy(logits, targets, comm=None, vocab_begin=0,
                                 ignore_index=-100):
    """Per-row losses for vocab-sharded (or full) logits."""
    return _VocabParallelCE.apply(logits.contiguous(), targets.contiguous(),
                                  comm, vocab_begin, ignore_index)


class ParallelCrossEntropy(torch.nn.Module):
    """Mean cross-entropy over non-ignored rows; drop-in for the reference's
    distributed_sparse_softmax_cross_entropy_with_logits wrapper."""

    def __init__(self, comm=None, vocab_begin=0, ignore_index=-100):
        super().__init__()
        self.comm = comm
        self.vocab_begin = vocab_begin
        self.ignore_index = ignore_index

    def forward(self, logits, targets):
        logits = logits.reshape(-1, logits.shape[-1])
        targets = targets.reshape(-1)
        losses = vocab_parallel_cross_entropy(
            logits, targets, self.comm, self.vocab_begin, self.ignore_index)
        valid = (targets != self.ignore_index).sum().clamp_min(1)
        return losses.sum() / valid
