"""Autograd-aware collectives.

Capability parity: /root/reference/epl/communicators/nccl_ops.py:37-124 —
the registered gradients of each collective: allreduce<->allreduce,
allgather<->reduce_scatter, reduce_scatter<->allgather, reduce<->broadcast,
alltoall(v)<->alltoall(v).  Implemented as torch.autograd.Functions over a
Communicator (GPU: RCCL core; CPU: gloo).
"""

import torch


class _AllReduce(torch.autograd.Function):
    @staticmethod
    def forward(ctx, inp, comm, op):
        ctx.comm = comm
        ctx.op = op
        out = inp.contiguous().clone()
        comm.all_reduce(out, op=op)
        return out

    @staticmethod
    def backward(ctx, grad):
        g = grad.contiguous().clone()
        ctx.comm.all_reduce(g, op=ctx.op)
        return g, None, None


def all_reduce(inp, comm, op="sum"):
    if comm.size == 1:
        return inp
    return _AllReduce.apply(inp, comm, op)


class _AllGather(torch.autograd.Function):
    """Concatenate along dim 0 across the group; backward reduce-scatters
    (reference: nccl_ops.py:53-62)."""

    @staticmethod
    def forward(ctx, inp, comm):
        ctx.comm = comm
        inp = inp.contiguous()
        out_shape = list(inp.shape)
        out_shape[0] *= comm.size
        out = inp.new_empty(out_shape)
        comm.all_gather(out, inp)
        return out

    @staticmethod
    def backward(ctx, grad):
        comm = ctx.comm
        grad = grad.contiguous()
        out_shape = list(grad.shape)
        out_shape[0] //= comm.size
        out = grad.new_empty(out_shape)
        comm.reduce_scatter(out, grad, op="sum")
        return out, None


def all_gather(inp, comm):
    if comm.size == 1:
        return inp
    return _AllGather.apply(inp, comm)


class _ReduceScatter(torch.autograd.Function):
    @staticmethod
    def forward(ctx, inp, comm, op):
        ctx.comm = comm
        inp = inp.contiguous()
        out_shape = list(inp.shape)
        out_shape[0] //= comm.size
        out = inp.new_empty(out_shape)
        comm.reduce_scatter(out, inp, op=op)
        return out

    @staticmethod
    def backward(ctx, grad):
        comm = ctx.comm
        grad = grad.contiguous()
        out_shape = list(grad.shape)
        out_shape[0] *= comm.size
        out = grad.new_empty(out_shape)
        comm.all_gather(out, grad)
        return out, None, None


def reduce_scatter(inp, comm, op="sum"):
    if comm.size == 1:
        return inp
    return _ReduceScatter.apply(inp, comm, op)


class _Broadcast(torch.autograd.Function):
    """Forward broadcast from root; backward reduces grads to root
    (reference: nccl_ops.py:81-97)."""

    @staticmethod
    def forward(ctx, inp, comm, root):
        ctx.comm = comm
        ctx.root = root
        out = inp.contiguous().clone()
        comm.broadcast(out, root=root)
        return out

    @staticmethod
    def backward(ctx, grad):
        g = grad.contiguous().clone()
        ctx.comm.reduce(g, root=ctx.root, op="sum")
        return g, None, None


def broadcast(inp, comm, root=0):
    if comm.size == 1:
        return inp
    return _Broadcast.apply(inp, comm, root)


class _AllToAll(torch.autograd.Function):
    """Equal-split all-to-all on dim 0; backward is all-to-all
    (reference: nccl_ops.py:99-124).  ``compress`` ('fp16'/'bf16')
    halves the wire bytes for fp32 payloads in BOTH directions
    (reference: parallel/ops.py:485-495 alltoall fp16 option)."""

    @staticmethod
    def forward(ctx, inp, comm, compress):
        ctx.comm = comm
        ctx.compress = compress
        inp = inp.contiguous()
        wire_dt = _wire_dtype(inp, compress)
        if wire_dt is not None:
            w = inp.to(wire_dt)
            o = w.new_empty(w.shape)
            comm.all_to_all_single(o, w)
            return o.to(inp.dtype)
        out = inp.new_empty(inp.shape)
        comm.all_to_all_single(out, inp)
        return out

    @staticmethod
    def backward(ctx, grad):
        grad = grad.contiguous()
        wire_dt = _wire_dtype(grad, ctx.compress)
        if wire_dt is not None:
            w = grad.to(wire_dt)
            o = w.new_empty(w.shape)
            ctx.comm.all_to_all_single(o, w)
            return o.to(grad.dtype), None, None
        out = grad.new_empty(grad.shape)
        ctx.comm.all_to_all_single(out, grad)
        return out, None, None


def _wire_dtype(t, compress):
    if not compress or t.dtype != torch.float32:
        return None
    return torch.float16 if compress == "fp16" else torch.bfloat16


def all_to_all(inp, comm, compress=""):
    if comm.size == 1:
        return inp
    return _AllToAll.apply(inp, comm, compress)


class _CopyToGroup(torch.autograd.Function):
    """Megatron 'f' operator: identity forward, all-reduce backward —
    marks the point where a replicated activation enters sharded
    compute (each shard's input-gradient is a partial sum)."""

    @staticmethod
    def forward(ctx, inp, comm):
        ctx.comm = comm
        return inp

    @staticmethod
    def backward(ctx, grad):
        g = grad.contiguous().clone()
        ctx.comm.all_reduce(g, op="sum")
        return g, None


def copy_to_group(inp, comm):
    if comm is None or comm.size == 1:
        return inp
    return _CopyToGroup.apply(inp, comm)


class _ReduceFromGroup(torch.autograd.Function):
    """Megatron 'g' operator: all-reduce forward, identity backward —
    for losses computed identically on every rank of the group."""

    @staticmethod
    def forward(ctx, inp, comm):
        out = inp.contiguous().clone()
        comm.all_reduce(out, op="sum")
        return out

    @staticmethod
    def backward(ctx, grad):
        return grad, None


def reduce_from_group(inp, comm):
    if comm is None or comm.size == 1:
        return inp
    return _ReduceFromGroup.apply(inp, comm)
