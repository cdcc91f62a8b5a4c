"""Sparse (IndexedSlices-style) gradient handling for
``nn.Embedding(sparse=True)`` / ``nn.EmbeddingBag(sparse=True)``.

Capability parity: /root/reference/epl/communicators/rewriters/
sparse_allreduce.py:39-173 — the reference allgathers (values, indices)
across the DP group instead of densifying, honoring
``communication.sparse_as_dense``.

MI355X redesign: sparse params are EXCLUDED from the flat arenas (a COO
gradient cannot live in a dense arena slice).  The wire path allgathers
the per-rank nnz counts, then the flattened indices and values
(all_gather_v — grouped ncclBroadcast over xGMI), rebuilds the summed
COO gradient, and applies it through a small dense-master AdamW/SGD side
step (mathematically identical to the dense path: a gathered+coalesced
sparse grad densifies to exactly the dense all-reduced grad).

``sparse_as_dense=True`` instead keeps the params in the arena and the
reducer densifies each grad as it adopts it (see FlatParamGroup
.adopt_grad) — full dense allreduce traffic, zero special casing.
"""

import torch

from easyparallellibrary_amd.utils.logging import get_logger

logger = get_logger()


def find_sparse_grad_params(modules):
    """Parameters that will receive sparse COO gradients."""
    import torch.nn as nn
    out = []
    seen = set()
    for root in modules:
        for m in root.modules():
            if isinstance(m, (nn.Embedding, nn.EmbeddingBag)) and m.sparse:
                if id(m.weight) not in seen and m.weight.requires_grad:
                    seen.add(id(m.weight))
                    out.append(m.weight)
    return out


class SparseGradHandler:
    """Gathers sparse grads across the DP group and steps the owning
    params with a dense fp32 master (AdamW semantics matching
    runtime/optim.FusedAdamW)."""

    def __init__(self, params, comm, reduce_method="mean", lr=1e-3,
                 betas=(0.9, 0.999), eps=1e-8, weight_decay=0.01):
        self.params = list(params)
        self.comm = comm
        self.reduce_method = reduce_method
        self.lr = lr
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.step_count = 0
        self.state = {}
        for p in self.params:
            master = p.data.detach().to(torch.float32).clone()
            self.state[id(p)] = {
                "master": master,
                "exp_avg": torch.zeros_like(master),
                "exp_avg_sq": torch.zeros_like(master),
            }

    # ---- comm ---------------------------------------------------------------
    def reduce(self):
        """Allgather (indices, values) across the group and install the
        summed (or averaged) global sparse gradient on every param."""
        w = self.comm.size if self.comm is not None else 1
        for p in self.params:
            g = p.grad
            if g is None:
                g = torch.sparse_coo_tensor(
                    torch.zeros((1, 0), dtype=torch.int64, device=p.device),
                    torch.zeros((0,) + p.shape[1:], dtype=p.dtype,
                                device=p.device), p.shape)
            if not g.is_sparse:
                raise RuntimeError(
                    "SparseGradHandler got a dense grad for a sparse "
                    "param; autograd layout changed")
            g = g.coalesce()
            if w > 1:
                idx, vals = g.indices(), g.values()
                nnz = idx.shape[1]
                counts = torch.empty(w, dtype=torch.int64, device=idx.device)
                self.comm.all_gather(
                    counts, torch.tensor([nnz], dtype=torch.int64,
                                         device=idx.device))
                row_elems = 1
                for d in p.shape[1:]:
                    row_elems *= int(d)
                idx_outs = [torch.empty(int(c), dtype=idx.dtype,
                                        device=idx.device)
                            for c in counts]
                val_outs = [torch.empty(int(c) * row_elems, dtype=vals.dtype,
                                        device=vals.device)
                            for c in counts]
                self.comm.all_gather_v(idx_outs, idx.reshape(-1).contiguous())
                self.comm.all_gather_v(val_outs,
                                       vals.reshape(-1).contiguous())
                all_idx = torch.cat(idx_outs).reshape(1, -1)
                all_vals = torch.cat(val_outs).reshape(
                    (-1,) + tuple(p.shape[1:]))
                g = torch.sparse_coo_tensor(all_idx, all_vals,
                                            p.shape).coalesce()
            if self.reduce_method == "mean" and w > 1:
                g = torch.sparse_coo_tensor(
                    g.indices(), g.values() / w, p.shape)
            p.grad = g

    def sqnorm(self):
        """||g||^2 of the post-reduce global sparse grads (for global
        grad-norm clipping; every rank in the group holds an identical
        copy, the caller divides by the copy count)."""
        total = 0.0
        dev = self.params[0].device if self.params else "cpu"
        acc = torch.zeros(1, dtype=torch.float32, device=dev)
        for p in self.params:
            if p.grad is not None:
                v = p.grad.coalesce().values()
                acc += v.float().pow(2).sum()
        return acc + total

    # ---- optimizer ----------------------------------------------------------
    def step(self, grad_scale=1.0):
        """Dense-master AdamW on the gathered gradient (identical math to
        the arena path — AdamW decays/updates every row each step, so the
        dense apply is the correct semantics; the sparsity saved wire
        traffic, not optimizer work)."""
        self.step_count += 1
        inv = 1.0 / grad_scale
        for p in self.params:
            st = self.state[id(p)]
            g = p.grad
            grad = (g.to_dense() if g is not None and g.is_sparse
                    else (g if g is not None else torch.zeros_like(p)))
            grad = grad.to(torch.float32)
            if inv != 1.0:
                grad = grad * inv
            m, v = st["exp_avg"], st["exp_avg_sq"]
            m.mul_(self.beta1).add_(grad, alpha=1 - self.beta1)
            v.mul_(self.beta2).addcmul_(grad, grad, value=1 - self.beta2)
            bc1 = 1 - self.beta1 ** self.step_count
            bc2 = 1 - self.beta2 ** self.step_count
            denom = (v / bc2).sqrt_().add_(self.eps)
            update = (m / bc1) / denom
            update.add_(st["master"], alpha=self.weight_decay)
            st["master"].add_(update, alpha=-self.lr)
            p.data.copy_(st["master"].to(p.dtype))

    def zero_grad(self):
        for p in self.params:
            p.grad = None

    def refresh_master(self):
        """Re-derive the fp32 masters from (possibly freshly loaded)
        params — mirrors FlatParamGroup.refresh_master's role in the
        checkpoint-load ordering (masters refresh BEFORE an optimizer
        load so a loaded exact master wins)."""
        for p in self.params:
            self.state[id(p)]["master"].copy_(p.data.to(torch.float32))

    # ---- checkpoint ---------------------------------------------------------
    def state_dict(self):
        return {
            "step": self.step_count,
            "lr": self.lr,
            "params": [
                {"master": self.state[id(p)]["master"],
                 "exp_avg": self.state[id(p)]["exp_avg"],
                 "exp_avg_sq": self.state[id(p)]["exp_avg_sq"]}
                for p in self.params
            ],
        }

    def load_state_dict(self, sd):
        self.step_count = sd["step"]
        self.lr = sd.get("lr", self.lr)
        for p, ps in zip(self.params, sd["params"]):
            st = self.state[id(p)]
            st["master"].copy_(ps["master"])
            st["exp_avg"].copy_(ps["exp_avg"])
            st["exp_avg_sq"].copy_(ps["exp_avg_sq"])
            p.data.copy_(st["master"].to(p.dtype))
