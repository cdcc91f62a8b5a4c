"""Flat-arena fused optimizers.

Capability parity: /root/reference/epl/ops/adam_weight_decay_optimizer.py
(AdamW :103-273) and the grouped-apply machinery of
runtime/optimizer_helper.py.

MI355X redesign: the optimizer state lives in flat fp32 arenas parallel to
the FlatParamGroup's master arena, so one step is ONE hand-written CDNA4
kernel launch per group (csrc/kernels/kernels.hip: fused_adamw / lamb
phases) — grad unscale and the bf16 parameter writeback are fused in.  On
CPU the same math runs as flat torch ops (numerics tests compare the HIP
kernel against this path).
"""

import math

import torch

from easyparallellibrary_amd.ops.dispatch import native_ext, use_native


class FusedAdamW:
    """AdamW over a list of FlatParamGroups."""

    def __init__(self, groups, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0.01):
        self.groups = list(groups)
        self.lr = lr
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.step_count = 0
        self._eager = False
        self._eager_scale = 1.0
        self._apply_stream = None
        for g in self.groups:
            g.state["exp_avg"] = torch.zeros_like(g.master_arena)
            g.state["exp_avg_sq"] = torch.zeros_like(g.master_arena)

    # ---- PreferBackwardOptimizer eager apply --------------------------------
    # (reference: strategies/scheduler.py:87-116 — the apply of a
    # parameter group starts as soon as its gradients are final, instead
    # of after the whole backward; here "group" = an arena bucket, and
    # the apply runs on a side stream fenced behind that bucket's comm.)
    def begin_eager(self, grad_scale):
        self.step_count += 1
        self._eager_scale = float(grad_scale)
        self._eager = True
        if self._apply_stream is None and torch.cuda.is_available():
            self._apply_stream = torch.cuda.Stream()

    def eager_apply(self, g, lo, hi, comm):
        """Apply AdamW to the arena slice [lo, hi) whose grads just
        became final (bucket reduced).  Called from the reducer's
        bucket callback during the final backward."""
        inv = 1.0 / self._eager_scale
        if use_native(g.master_arena):
            with torch.cuda.stream(self._apply_stream):
                comm.join()   # fence apply stream behind the comm stream
                native_ext().fused_adamw(
                    g.master_arena[lo:hi],
                    g.param_arena[lo:hi]
                    if g.param_arena.dtype == torch.bfloat16 else None,
                    g.grad_arena[lo:hi], g.state["exp_avg"][lo:hi],
                    g.state["exp_avg_sq"][lo:hi],
                    self.lr, self.beta1, self.beta2, self.eps,
                    self.weight_decay, self.step_count, inv)
        else:
            self._step_torch_slice(g, lo, hi, inv)

    def _step_torch_slice(self, g, lo, hi, inv_scale):
        grad = g.grad_arena[lo:hi].to(torch.float32)
        if inv_scale != 1.0:
            grad = grad * inv_scale
        m = g.state["exp_avg"][lo:hi]
        v = g.state["exp_avg_sq"][lo:hi]
        m.mul_(self.beta1).add_(grad, alpha=1 - self.beta1)
        v.mul_(self.beta2).addcmul_(grad, grad, value=1 - self.beta2)
        bc1 = 1 - self.beta1 ** self.step_count
        bc2 = 1 - self.beta2 ** self.step_count
        denom = (v / bc2).sqrt_().add_(self.eps)
        update = (m / bc1) / denom
        update.add_(g.master_arena[lo:hi], alpha=self.weight_decay)
        g.master_arena[lo:hi].add_(update, alpha=-self.lr)
        if g.master_arena is not g.param_arena:
            g.param_arena[lo:hi].copy_(
                g.master_arena[lo:hi].to(g.param_arena.dtype))

    def step(self, grad_scale=1.0):
        if self._eager:
            # buckets were applied during backward; just fence the next
            # forward behind the apply stream
            self._eager = False
            if self._apply_stream is not None:
                torch.cuda.current_stream().wait_stream(self._apply_stream)
            return
        self.step_count += 1
        inv_scale = 1.0 / grad_scale
        from easyparallellibrary_amd.env import Env
        napply = max(1, Env.get().config.optimizer.num_apply_group)
        for g in self.groups:
            if use_native(g.master_arena):
                # grouped apply (reference: runtime/optimizer_helper.py
                # :75-128): the flat arena makes this a chunked launch —
                # one fused kernel per apply group
                n = g.master_arena.numel()
                chunk = (n + napply - 1) // napply
                chunk = (chunk + 127) // 128 * 128
                for lo in range(0, n, chunk):
                    hi = min(lo + chunk, n)
                    native_ext().fused_adamw(
                        g.master_arena[lo:hi],
                        g.param_arena[lo:hi]
                        if g.param_arena.dtype == torch.bfloat16 else None,
                        g.grad_arena[lo:hi], g.state["exp_avg"][lo:hi],
                        g.state["exp_avg_sq"][lo:hi],
                        self.lr, self.beta1, self.beta2, self.eps,
                        self.weight_decay, self.step_count, inv_scale)
            else:
                self._step_torch(g, inv_scale)

    def _step_torch(self, g, inv_scale):
        grad = g.grad_arena.to(torch.float32)
        if inv_scale != 1.0:
            grad = grad * inv_scale
        m, v = g.state["exp_avg"], g.state["exp_avg_sq"]
        m.mul_(self.beta1).add_(grad, alpha=1 - self.beta1)
        v.mul_(self.beta2).addcmul_(grad, grad, value=1 - self.beta2)
        bc1 = 1 - self.beta1 ** self.step_count
        bc2 = 1 - self.beta2 ** self.step_count
        denom = (v / bc2).sqrt_().add_(self.eps)
        update = (m / bc1) / denom
        update.add_(g.master_arena, alpha=self.weight_decay)
        g.master_arena.add_(update, alpha=-self.lr)
        g.sync_master_to_params()

    def zero_grad(self):
        for g in self.groups:
            g.zero_grad()

    def state_dict(self):
        return {
            "step": self.step_count,
            "lr": self.lr,
            "groups": [
                {"exp_avg": g.state["exp_avg"],
                 "exp_avg_sq": g.state["exp_avg_sq"],
                 "master": g.master_arena}
                for g in self.groups
            ],
        }

    def load_state_dict(self, sd):
        self.step_count = sd["step"]
        self.lr = sd.get("lr", self.lr)
        for g, gs in zip(self.groups, sd["groups"]):
            g.state["exp_avg"].copy_(gs["exp_avg"])
            g.state["exp_avg_sq"].copy_(gs["exp_avg_sq"])
            g.master_arena.copy_(gs["master"])
            g.sync_master_to_params()


class FusedLAMB:
    """LAMB over FlatParamGroups: per-parameter trust ratio
    (chunk = parameter extent in the flat arena)."""

    def __init__(self, groups, lr=1e-3, betas=(0.9, 0.999), eps=1e-6,
                 weight_decay=0.01):
        self.groups = list(groups)
        self.lr = lr
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.step_count = 0
        for g in self.groups:
            g.state["exp_avg"] = torch.zeros_like(g.master_arena)
            g.state["exp_avg_sq"] = torch.zeros_like(g.master_arena)
            g.state["lamb_update"] = torch.zeros_like(g.master_arena)
            nchunks = len(g.ordered)
            # per-8-element chunk index map (kernel indexes chunk_of[i>>3])
            cmap = torch.empty((g.total + 7) // 8, dtype=torch.int32)
            for ci, (p, off) in enumerate(zip(g.ordered, g.offsets)):
                end = off + p.numel()
                cmap[off // 8:(end + 7) // 8] = ci
            g.state["lamb_chunk_of"] = cmap.to(g.master_arena.device)
            g.state["lamb_nchunks"] = nchunks

    def step(self, grad_scale=1.0):
        self.step_count += 1
        inv_scale = 1.0 / grad_scale
        for g in self.groups:
            n = g.state["lamb_nchunks"]
            dev = g.master_arena.device
            wsq = torch.zeros(n, dtype=torch.float32, device=dev)
            usq = torch.zeros(n, dtype=torch.float32, device=dev)
            if use_native(g.master_arena):
                ext = native_ext()
                ext.lamb_phase1(
                    g.master_arena, g.grad_arena, g.state["exp_avg"],
                    g.state["exp_avg_sq"], g.state["lamb_update"],
                    g.state["lamb_chunk_of"], wsq, usq, self.beta1,
                    self.beta2, self.eps, self.weight_decay, self.step_count,
                    inv_scale)
                ratio = torch.where(
                    (wsq > 0) & (usq > 0),
                    torch.sqrt(wsq) / torch.sqrt(usq).clamp_min(1e-12),
                    torch.ones_like(wsq))
                ext.lamb_phase2(
                    g.master_arena,
                    g.param_arena if g.param_arena.dtype == torch.bfloat16
                    else None,
                    g.state["lamb_update"], g.state["lamb_chunk_of"], ratio,
                    self.lr)
            else:
                self._step_torch(g, inv_scale)

    def _step_torch(self, g, inv_scale):
        grad = g.grad_arena.to(torch.float32)
        if inv_scale != 1.0:
            grad = grad * inv_scale
        m, v = g.state["exp_avg"], g.state["exp_avg_sq"]
        m.mul_(self.beta1).add_(grad, alpha=1 - self.beta1)
        v.mul_(self.beta2).addcmul_(grad, grad, value=1 - self.beta2)
        bc1 = 1 - self.beta1 ** self.step_count
        bc2 = 1 - self.beta2 ** self.step_count
        update = (m / bc1) / ((v / bc2).sqrt() + self.eps)
        update.add_(g.master_arena, alpha=self.weight_decay)
        for p, off in zip(g.ordered, g.offsets):
            n = p.numel()
            w = g.master_arena[off:off + n]
            u = update[off:off + n]
            wn = w.norm()
            un = u.norm()
            ratio = (wn / un.clamp_min(1e-12)).item() if wn > 0 and un > 0 \
                else 1.0
            w.add_(u, alpha=-self.lr * ratio)
        g.sync_master_to_params()

    def zero_grad(self):
        for g in self.groups:
            g.zero_grad()


OPTIMIZERS = {"adamw": FusedAdamW, "lamb": FusedLAMB}
