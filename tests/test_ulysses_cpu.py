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


def _ulysses_engine_worker(rank, world):
    """Engine-integrated Ulysses SP (mirror of the ring-attention
    engine test): sequence-sharded data, replicated params, DP-averaged
    grads across the SP group."""
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.comm.backend import create_communicator
    from easyparallellibrary_amd.env import Env
    from easyparallellibrary_amd.ops.ulysses import UlyssesSelfAttention

    epl.init()
    Env.get().get_or_create_process_group()
    sp = (create_communicator("sp_ul_eng", list(range(world)))
          if world > 1 else None)
    torch.manual_seed(90)

    class LM(nn.Module):
        def __init__(self):
            super().__init__()
            self.emb = nn.Embedding(64, H)
            self.attn = UlyssesSelfAttention(H, HEADS, comm=sp,
                                             causal=True)
            self.head = nn.Linear(H, 64)

        def forward(self, ids):
            return self.head(self.attn(self.emb(ids)))

    def lm_loss(logits, targets):
        return nn.functional.cross_entropy(logits.reshape(-1, 64),
                                           targets)

    with epl.replicate(1):
        model = LM()
    engine = epl.Engine(model, loss_fn=lm_loss, optimizer="adamw",
                        lr=1e-2)
    torch.manual_seed(91)
    ids = torch.randint(0, 64, (2, S))
    tgt = torch.randint(0, 64, (2, S))
    sl = S // world
    lo = rank * sl
    out = []
    for _ in range(3):
        loss = engine.train_step(ids[:, lo:lo + sl],
                                 tgt[:, lo:lo + sl].reshape(-1))
        out.append(float(engine.all_reduce_metric(loss)))
    return out


def test_ulysses_sp_engine_matches_serial():
    serial = _ulysses_engine_worker(0, 1)
    from easyparallellibrary_amd.env import Env
    from easyparallellibrary_amd.parallel import hooks
    hooks.remove_hooks()
    Env._instance = None
    sp2 = run_multiprocess(_ulysses_engine_worker, world=2)
    assert sp2[0] == sp2[1]
    for a, b in zip(serial, sp2[0]):
        assert abs(a - b) < 1e-5, (serial, sp2[0])
