"""Ring attention (context parallelism) vs serial full attention —
beyond-parity extension #2 (see ops/ring_attention.py)."""

import math

import torch
import torch.nn as nn

from tests.utils import run_multiprocess


def _ref_attention(q, k, v, causal, scale):
    s = (q.float() @ k.float().transpose(-1, -2)) * scale
    if causal:
        n = s.shape[-1]
        mask = torch.triu(torch.ones(n, n, dtype=torch.bool), 1)
        s = s.masked_fill(mask, float("-inf"))
    return (s.softmax(-1) @ v.float()).to(q.dtype)


def _ring_worker(rank, world, causal):
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.comm.backend import create_communicator
    from easyparallellibrary_amd.env import Env
    from easyparallellibrary_amd.ops.ring_attention import ring_attention
    epl.init()
    Env.get().get_or_create_process_group()
    comm = create_communicator("ring", list(range(world)))
    torch.manual_seed(77)
    b, h, s, d = 2, 3, 32, 16      # h < world-compatible (ring keeps heads)
    q = torch.randn(b, h, s, d)
    k = torch.randn(b, h, s, d)
    v = torch.randn(b, h, s, d)
    scale = 1.0 / math.sqrt(d)
    sl = s // world
    lo = rank * sl
    ql = q[:, :, lo:lo + sl].clone().requires_grad_(True)
    kl = k[:, :, lo:lo + sl].clone().requires_grad_(True)
    vl = v[:, :, lo:lo + sl].clone().requires_grad_(True)
    out = ring_attention(ql, kl, vl, comm, causal=causal, scale=scale)
    # identical per-position weights in local and full numbering
    wfull = torch.sin(torch.arange(q.numel()).reshape(q.shape).float())
    wloc = wfull[:, :, lo:lo + sl]
    (out * wloc).sum().backward()

    qf = q.clone().requires_grad_(True)
    kf = k.clone().requires_grad_(True)
    vf = v.clone().requires_grad_(True)
    ref = _ref_attention(qf, kf, vf, causal, scale)
    w_masked = torch.zeros_like(wfull)
    w_masked[:, :, lo:lo + sl] = wloc
    # serial loss restricted to MY block's outputs gives MY dq
    (ref * w_masked).sum().backward()
    fwd_err = (out - ref[:, :, lo:lo + sl]).abs().max().item()
    dq_err = (ql.grad - qf.grad[:, :, lo:lo + sl]).abs().max().item()
    return fwd_err, dq_err


def test_ring_attention_matches_serial():
    for causal in (False, True):
        res = run_multiprocess(_ring_worker, world=2, args=(causal,))
        for fwd_err, dq_err in res:
            assert fwd_err < 1e-5, (causal, fwd_err)
            assert dq_err < 1e-4, (causal, dq_err)


def test_ring_attention_world4_causal():
    """More hops (3 rotations), causal masking across blocks."""
    res = run_multiprocess(_ring_worker, world=4, args=(True,),
                           timeout=240)
    for fwd_err, dq_err in res:
        assert fwd_err < 1e-5, fwd_err
        assert dq_err < 1e-4, dq_err


def test_ring_module_single_rank_matches_plain():
    """world=1 degenerate: RingSelfAttention == plain attention."""
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.ops.ring_attention import (
        RingSelfAttention)
    epl.init()
    torch.manual_seed(3)
    attn = RingSelfAttention(32, 4, comm=None, causal=True)
    x = torch.randn(2, 16, 32)
    y = attn(x)
    # reference through the same weights
    b, s, h = x.shape
    qkv = attn.qkv(x).reshape(b, s, 3, 4, 8)
    q, k, v = (qkv[:, :, i].transpose(1, 2) for i in range(3))
    ref = attn.proj(_ref_attention(q, k, v, True, 8 ** -0.5)
                    .transpose(1, 2).reshape(b, s, h))
    assert (y - ref).abs().max().item() < 1e-5


class _TinyLM(nn.Module):
    def __init__(self, vocab, hidden, heads, comm):
        super().__init__()
        from easyparallellibrary_amd.ops.ring_attention import (
            RingSelfAttention)
        self.emb = nn.Embedding(vocab, hidden)
        self.attn = RingSelfAttention(hidden, heads, comm=comm,
                                      causal=True)
        self.head = nn.Linear(hidden, vocab)

    def forward(self, ids_and_pos):
        ids, pos = ids_and_pos
        x = self.emb(ids) + 0.01 * pos.unsqueeze(-1)
        return self.head(self.attn(x))


def _sp_engine_worker(rank, world):
    """Engine-integrated sequence parallelism: each rank trains on its
    sequence shard; params replicate and grads DP-average across the SP
    group (the mean of per-shard means == the global mean for equal
    shards)."""
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.comm.backend import create_communicator
    from easyparallellibrary_amd.env import Env
    epl.init()
    Env.get().get_or_create_process_group()
    sp = (create_communicator("sp_eng", list(range(world)))
          if world > 1 else None)
    torch.manual_seed(90)
    with epl.replicate(1):
        model = _TinyLM(64, 32, 4, sp)
    def lm_loss(logits, targets):
        return nn.functional.cross_entropy(logits.reshape(-1, 64),
                                           targets)

    engine = epl.Engine(model, loss_fn=lm_loss, optimizer="adamw",
                        lr=1e-2)
    torch.manual_seed(91)
    s = 32
    ids = torch.randint(0, 64, (2, s))
    tgt = torch.randint(0, 64, (2, s))
    pos = torch.arange(s, dtype=torch.float32).expand(2, s)
    sl = s // world
    lo = rank * sl
    losses = []
    for _ in range(3):
        loss = engine.train_step(
            (ids[:, lo:lo + sl], pos[:, lo:lo + sl]),
            tgt[:, lo:lo + sl].reshape(-1))
        losses.append(float(engine.all_reduce_metric(loss)))
    return losses


def test_ring_sp_engine_matches_serial():
    serial = _sp_engine_worker(0, 1)
    from easyparallellibrary_amd.env import Env
    from easyparallellibrary_amd.parallel import hooks
    hooks.remove_hooks()
    Env._instance = None
    sp2 = run_multiprocess(_sp_engine_worker, world=2)
    assert sp2[0] == sp2[1]
    for a, b in zip(serial, sp2[0]):
        assert abs(a - b) < 1e-5, (serial, sp2[0])
