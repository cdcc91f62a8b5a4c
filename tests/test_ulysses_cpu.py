"""Ulysses sequence parallelism on CPU/gloo: SP2 attention over sharded
sequences must match the serial module bit-for-bit (fp32), forward and
backward."""

import torch
import torch.nn as nn

from tests.utils import run_multiprocess

B, S, H, HEADS = 2, 16, 32, 4


def _worker(rank, world):
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.comm.backend import create_communicator
    from easyparallellibrary_amd.env import Env
    from easyparallellibrary_amd.ops.ulysses import UlyssesSelfAttention

    epl.init()
    Env.get().get_or_create_process_group()
    comm = create_communicator("sp", list(range(world)))
    torch.manual_seed(7)
    attn = UlyssesSelfAttention(H, HEADS, comm=comm, causal=True)
    torch.manual_seed(8)
    x = torch.randn(B, S, H, requires_grad=True)
    sl = S // world
    xs = x[:, rank * sl:(rank + 1) * sl].detach().requires_grad_(True)
    out = attn(xs)
    out.sum().backward()
    return (out.detach(), xs.grad,
            attn.qkv.weight.grad.clone(), attn.proj.weight.grad.clone())


def test_ulysses_sp2_matches_serial():
    results = run_multiprocess(_worker, world=2)
    # serial reference
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.ops.ulysses import UlyssesSelfAttention
    epl.init()
    torch.manual_seed(7)
    attn = UlyssesSelfAttention(H, HEADS, comm=None, causal=True)
    torch.manual_seed(8)
    x = torch.randn(B, S, H, requires_grad=True)
    out = attn(x)
    out.sum().backward()

    sl = S // 2
    for r, (o, xg, qkvg, projg) in enumerate(results):
        assert torch.allclose(o, out[:, r * sl:(r + 1) * sl].detach(),
                              atol=1e-5), r
        assert torch.allclose(xg, x.grad[:, r * sl:(r + 1) * sl],
                              atol=1e-5), r
        # weight grads: serial = sum of the per-rank shard grads
    total_qkv = results[0][2] + results[1][2]
    assert torch.allclose(total_qkv, attn.qkv.weight.grad, atol=1e-5)
    total_proj = results[0][3] + results[1][3]
    assert torch.allclose(total_proj, attn.proj.weight.grad, atol=1e-5)
