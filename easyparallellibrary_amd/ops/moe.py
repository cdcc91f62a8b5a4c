"""Mixture-of-Experts with expert parallelism (all-to-all dispatch/combine).

Capability parity: /root/reference/epl/parallel/hooks.py:758-794 (einsum
hook inserting all-to-all before the first and after the third einsum
around the expert weights) + parallel/ops.py:485-495 (alltoall with
optional fp16 compression) + csrc nccl_all_to_all.cc.

MI355X redesign: an explicit MoE layer — top-1/top-2 gating with capacity,
dispatch as ONE equal-split all-to-all over xGMI, local expert FFN batched
as a single bmm over the rank's experts, combine with the reverse
all-to-all.  The a2a autograd pair (a2a <-> a2a) lives in
comm/functional.py.
"""

import torch
import torch.nn as nn
import torch.nn.functional as F

from easyparallellibrary_amd.comm import functional
from easyparallellibrary_amd.ops.dispatch import native_ext, use_native


def _moe_bmm():
    import os
    return os.environ.get("EPL_MOE_BMM", "1") == "1"


def _moe_native_dispatch():
    import os
    return os.environ.get("EPL_MOE_NATIVE_DISPATCH", "1") == "1"


class _MoEDispatch(torch.autograd.Function):
    """Fused token dispatch (csrc/kernels/moe.hip): one wave copies one
    hidden row per kept assignment; over-capacity entries are skipped
    INLINE (no boolean compaction, no nonzero device->host sync).
    Backward gathers each token's k slot-gradients (dropped -> zero)."""

    @staticmethod
    def forward(ctx, x, fe, pos, ft, inv, k, capacity, num_experts):
        disp = x.new_zeros(num_experts, capacity, x.shape[-1])
        native_ext().moe_dispatch_fwd(disp, x, fe, pos, ft, capacity)
        ctx.save_for_backward(fe, pos, inv)
        ctx.k = k
        ctx.cap = capacity
        ctx.n = x.shape[0]
        return disp

    @staticmethod
    def backward(ctx, ddisp):
        fe, pos, inv = ctx.saved_tensors
        dx = ddisp.new_empty(ctx.n, ddisp.shape[-1])
        native_ext().moe_dispatch_bwd(dx, ddisp.contiguous(), fe, pos,
                                      inv, ctx.k, ctx.cap)
        return dx, None, None, None, None, None, None, None


class _MoECombine(torch.autograd.Function):
    """Fused weighted combine: out[t] = sum_j fw * h[slot_j(t)] with
    fp32 accumulation; backward writes dh rows (unique slots, no
    atomics) and per-assignment dfw dot products."""

    @staticmethod
    def forward(ctx, h, fw, fe, pos, ft, inv, k, capacity, n_tokens):
        h = h.contiguous()
        fwf = fw.detach().float().contiguous()
        out = h.new_empty(n_tokens, h.shape[-1])
        native_ext().moe_combine_fwd(out, h, fwf, fe, pos, inv, k,
                                     capacity)
        ctx.save_for_backward(h, fwf, fe, pos, ft)
        ctx.cap = capacity
        return out

    @staticmethod
    def backward(ctx, dout):
        h, fwf, fe, pos, ft = ctx.saved_tensors
        dh = torch.zeros_like(h)
        dfw = torch.empty_like(fwf)
        native_ext().moe_combine_bwd(dh, dfw, dout.contiguous(), h, fwf,
                                     fe, pos, ft, ctx.cap)
        return dh, dfw, None, None, None, None, None, None, None


class _BatchedExpertLinear(torch.autograd.Function):
    """Per-expert batched GEMM with a hand-written backward.

    torch.bmm's own autograd backward memory-faults on
    ROCm 7.0.x/gfx950 (still reproduced on 7.0.2 —
    gpurun_out/r2_moe_bmm2.txt, repro tests/moe_bisect_gpu.py bmm2).
    Issuing the backward GEMMs OURSELVES as forward bmm calls (with
    transposed-view operands — probed fault-free and faster than
    transpose copies, tests/moe_bmm_layout_probe_gpu.py) dodges it
    while keeping one grouped GEMM launch per matmul instead of the
    per-expert 2D loop."""

    @staticmethod
    def forward(ctx, x, w):
        ctx.save_for_backward(x, w)
        return torch.bmm(x, w)

    @staticmethod
    def backward(ctx, go):
        x, w = ctx.saved_tensors
        go = go.contiguous()
        # transposed VIEWS where the layout is safe (probed per-GEMM in
        # tests/moe_bmm_layout_probe_gpu.py): the ONE faulting hipBLASLt
        # config on this stack is NT with a narrow inner dim
        # ([E,C,1024] @ [E,1024,4096]-view) — exactly gx of the second
        # expert linear; that operand alone gets a contiguous copy
        # (~175 us/layer) while the other three GEMMs ride views
        wt = w.transpose(1, 2)
        if wt.shape[-1] > wt.shape[-2]:   # widening NT: the faulting one
            wt = wt.contiguous()
        gx = torch.bmm(go, wt)
        gw = torch.bmm(x.transpose(1, 2), go)
        return gx, gw


class ExpertParallelMLP(nn.Module):
    """num_experts split across the comm group; each rank holds
    num_experts // world experts (reference tests use the same layout,
    tests/split_test.py:30-90)."""

    def __init__(self, hidden, ffn_hidden, num_experts, comm=None, top_k=2,
                 capacity_factor=1.25):
        super().__init__()
        self.comm = comm
        self.world = comm.size if comm is not None else 1
        assert num_experts % self.world == 0, \
            "num_experts must divide the split degree"
        self.num_experts = num_experts
        self.ffn_hidden = ffn_hidden
        self.local_experts = num_experts // self.world
        self.hidden = hidden
        self.top_k = top_k
        self.capacity_factor = capacity_factor
        self.gate = nn.Linear(hidden, num_experts, bias=False)
        self.w1 = nn.Parameter(
            torch.empty(self.local_experts, hidden, ffn_hidden))
        self.w2 = nn.Parameter(
            torch.empty(self.local_experts, ffn_hidden, hidden))
        nn.init.normal_(self.w1, std=0.02)
        nn.init.normal_(self.w2, std=0.02)
        if self.world > 1:
            self.w1._epl_shard_dim = 0
            self.w2._epl_shard_dim = 0

    def set_comm(self, comm):
        """Called by the engine's split transform: shard the expert weights
        constructed with comm=None (full expert set) down to this rank's
        slice (reference: expert weights inside epl.split, hooks.py:758)."""
        if comm is None or comm.size == 1:
            self.comm = comm
            return
        assert self.world == 1, "expert weights already sharded"
        self.comm = comm
        self.world = comm.size
        assert self.num_experts % self.world == 0
        self.local_experts = self.num_experts // self.world
        shard = max(comm.rank, 0)
        lo = shard * self.local_experts
        hi = lo + self.local_experts
        with torch.no_grad():
            self.w1 = nn.Parameter(self.w1[lo:hi].clone())
            self.w2 = nn.Parameter(self.w2[lo:hi].clone())
        self.w1._epl_shard_dim = 0
        self.w2._epl_shard_dim = 0

    def _wire_compression(self):
        # reference parallel/ops.py:485-495: optional fp16 all-to-all;
        # only meaningful for fp32 activations (bf16 ships bf16 anyway)
        from easyparallellibrary_amd.env import Env
        return Env.get().config.communication.compression

    def forward(self, x):
        orig_shape = x.shape
        x = x.reshape(-1, self.hidden)
        n_tokens = x.shape[0]
        logits = self.gate(x).float()
        probs = logits.softmax(dim=-1)
        topv, topi = probs.topk(self.top_k, dim=-1)
        topv = topv / topv.sum(dim=-1, keepdim=True)

        capacity = max(
            1, int(self.capacity_factor * n_tokens * self.top_k /
                   self.num_experts))
        flat_e = topi.reshape(-1)
        flat_t = (torch.arange(n_tokens, device=x.device)
                  .repeat_interleave(self.top_k))
        flat_w = topv.reshape(-1)
        # slot assignment per expert (ordered, capacity-dropped)
        order = torch.argsort(flat_e, stable=True)
        fe, ft, fw = flat_e[order], flat_t[order], flat_w[order]
        # position within expert via segmented arange.  scatter_add of
        # ones instead of bincount: bincount infers its output size on
        # the host (device sync), which both stalls the step and is
        # hipGraph-capture-unsupported; integer atomics make the
        # scatter_add exact and order-independent.
        counts = torch.zeros(self.num_experts, dtype=torch.long,
                             device=x.device)
        counts.scatter_add_(0, fe, torch.ones_like(fe))
        seg_start = torch.nn.functional.pad(counts.cumsum(0), (1, 0))[:-1]
        pos_in_e = torch.arange(fe.numel(), device=x.device) - seg_start[fe]
        native = (_moe_native_dispatch() and use_native(x)
                  and x.dtype == torch.bfloat16 and self.hidden % 8 == 0)
        inv = None
        if native:
            # fused dispatch kernel: full assignment list, in-kernel
            # capacity skip — no boolean select / nonzero sync
            m = fe.numel()
            inv = torch.empty_like(order)
            inv[order] = torch.arange(m, device=x.device)
            dispatched = _MoEDispatch.apply(
                x, fe, pos_in_e, ft, inv, self.top_k, capacity,
                self.num_experts)
        else:
            keep = pos_in_e < capacity
            fe, ft, fw, pos_in_e = (fe[keep], ft[keep], fw[keep],
                                    pos_in_e[keep])
            # dispatch tensor: [num_experts, capacity, hidden]
            dispatched = x.new_zeros(self.num_experts, capacity,
                                     self.hidden)
            dispatched[fe, pos_in_e] = x[ft]

        # all-to-all: [world, local_experts*capacity, hidden]
        d = dispatched.reshape(self.world,
                               self.local_experts * capacity, self.hidden)
        if self.comm is not None and self.world > 1:
            d = functional.all_to_all(d.contiguous(), self.comm,
                                      compress=self._wire_compression())
        # now d[w] = tokens sent by rank w for MY local experts
        d = d.reshape(self.world, self.local_experts, capacity, self.hidden)
        d = d.transpose(0, 1).reshape(self.local_experts,
                                      self.world * capacity, self.hidden)
        # expert FFN: batched GEMM (one hipBLASLt grouped call per
        # matmul, hand-written backward above) by default; EPL_MOE_BMM=0
        # falls back to the per-expert 2D-GEMM loop
        if _moe_bmm():
            he = F.gelu(_BatchedExpertLinear.apply(d.contiguous(), self.w1))
            h = _BatchedExpertLinear.apply(he, self.w2)
        else:
            outs = []
            for e in range(self.local_experts):
                he = F.gelu(d[e] @ self.w1[e])
                outs.append(he @ self.w2[e])
            h = torch.stack(outs)
        h = h.reshape(self.local_experts, self.world, capacity, self.hidden)
        h = h.transpose(0, 1).reshape(
            self.world, self.local_experts * capacity, self.hidden)
        if self.comm is not None and self.world > 1:
            h = functional.all_to_all(h.contiguous(), self.comm,
                                      compress=self._wire_compression())
        h = h.reshape(self.num_experts, capacity, self.hidden)

        if native:
            out = _MoECombine.apply(h, fw, fe, pos_in_e, ft, inv,
                                    self.top_k, capacity, n_tokens)
        else:
            out = x.new_zeros(n_tokens, self.hidden)
            out.index_add_(0, ft,
                           h[fe, pos_in_e] * fw.unsqueeze(-1).to(h.dtype))
        return out.reshape(orig_shape)
