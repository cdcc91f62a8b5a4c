"""Sparse (IndexedSlices-style) embedding gradients — parity with the
reference's rewriters/sparse_allreduce.py:39-173 and the
communication.sparse_as_dense knob (config.py:81).

The gather-wire path (sparse_as_dense=False, default) must train a
``nn.Embedding(sparse=True)`` model to the SAME trajectory as the dense
embedding under DP — the gathered+coalesced sparse grad densifies to
exactly the dense all-reduced grad, and the side AdamW matches the
fused arena AdamW math."""

import torch
import torch.nn as nn

from tests.utils import run_multiprocess


class TinyLM(nn.Module):
    def __init__(self, vocab=50, hidden=16, sparse=True):
        super().__init__()
        self.emb = nn.Embedding(vocab, hidden, sparse=sparse)
        self.fc = nn.Linear(hidden, vocab)

    def forward(self, ids):
        return self.fc(self.emb(ids)).mean(dim=1)


def _ce(out, tgt):
    return nn.functional.cross_entropy(out, tgt)


def _train(rank, world, sparse, sparse_as_dense):
    import easyparallellibrary_amd as epl
    epl.init({"communication.sparse_as_dense": sparse_as_dense})
    torch.manual_seed(42)  # same init everywhere; bcast is a no-op check
    with epl.replicate(device_count=1):
        model = TinyLM(sparse=sparse)
    engine = epl.Engine(model, loss_fn=_ce, optimizer="adamw", lr=1e-2)
    torch.manual_seed(100 + rank)  # DIFFERENT data per rank
    losses = []
    for _ in range(4):
        ids = torch.randint(0, 50, (6, 5))
        tgt = torch.randint(0, 50, (6,))
        losses.append(float(engine.train_step(ids, tgt)))
    emb_w = model.emb.weight.detach().clone()
    fc_w = model.fc.weight.detach().clone()
    return losses, emb_w, fc_w


def _worker_gather(rank, world):
    return _train(rank, world, sparse=True, sparse_as_dense=False)


def _worker_dense_embedding(rank, world):
    return _train(rank, world, sparse=False, sparse_as_dense=False)


def _worker_sparse_as_dense(rank, world):
    return _train(rank, world, sparse=True, sparse_as_dense=True)


def test_sparse_gather_matches_dense_dp():
    got = run_multiprocess(_worker_gather, world=2)
    ref = run_multiprocess(_worker_dense_embedding, world=2)
    for r in range(2):
        gl, gw, gf = got[r]
        rl, rw, rf = ref[r]
        assert torch.allclose(gw, rw, atol=2e-5), (gw - rw).abs().max()
        assert torch.allclose(gf, rf, atol=2e-5)
        for a, b in zip(gl, rl):
            assert abs(a - b) < 1e-5, (gl, rl)
    # ranks stayed in sync
    assert torch.allclose(got[0][1], got[1][1], atol=0, rtol=0)


def test_sparse_as_dense_matches_dense_dp():
    got = run_multiprocess(_worker_sparse_as_dense, world=2)
    ref = run_multiprocess(_worker_dense_embedding, world=2)
    for r in range(2):
        assert torch.allclose(got[r][1], ref[r][1], atol=2e-5)
        for a, b in zip(got[r][0], ref[r][0]):
            assert abs(a - b) < 1e-5


def test_sparse_serial_single_rank():
    # world=1: no comm; the side optimizer must still step the embedding
    got = run_multiprocess(_worker_gather, world=1)
    losses = got[0][0]
    assert losses[-1] < losses[0]


def _worker_zero_raises(rank, world):
    import easyparallellibrary_amd as epl
    epl.init({"zero.level": "v1",
              "communication.sparse_as_dense": False})
    with epl.replicate(device_count=1):
        model = TinyLM(sparse=True)
    try:
        epl.Engine(model, loss_fn=_ce)
    except ValueError as e:
        return "raised: {}".format(e)
    return "no error"


def test_sparse_with_zero_raises():
    out = run_multiprocess(_worker_zero_raises, world=1)
    assert out[0].startswith("raised"), out


def _worker_ckpt(rank, world, path):
    import easyparallellibrary_amd as epl
    epl.init()
    torch.manual_seed(9)
    with epl.replicate(device_count=1):
        model = TinyLM(sparse=True)
    engine = epl.Engine(model, loss_fn=_ce, optimizer="adamw", lr=1e-2)
    ids = torch.randint(0, 50, (6, 5))
    tgt = torch.randint(0, 50, (6,))
    engine.train_step(ids, tgt)
    engine.save_checkpoint(path)
    l_next = float(engine.train_step(ids, tgt))

    torch.manual_seed(77)  # fresh, different init
    with epl.replicate(device_count=1):
        model2 = TinyLM(sparse=True)
    engine2 = epl.Engine(model2, loss_fn=_ce, optimizer="adamw",
                         lr=1e-2)
    engine2.load_checkpoint(path)
    l_resume = float(engine2.train_step(ids, tgt))
    return l_next, l_resume


def test_sparse_checkpoint_roundtrip(tmp_path):
    out = run_multiprocess(_worker_ckpt, world=1, args=(str(tmp_path),))
    l_next, l_resume = out[0]
    assert abs(l_next - l_resume) < 1e-6, (l_next, l_resume)


def _worker_ckpt_no_opt(rank, world, path):
    """load_optimizer=False must refresh the sparse fp32 master from
    the loaded params (regression: a stale master overwrote the loaded
    embedding on the first post-load step)."""
    import easyparallellibrary_amd as epl
    epl.init()
    torch.manual_seed(21)
    with epl.replicate(device_count=1):
        model = TinyLM(sparse=True)
    engine = epl.Engine(model, loss_fn=_ce, optimizer="adamw", lr=1e-2)
    ids = torch.randint(0, 50, (6, 5))
    tgt = torch.randint(0, 50, (6,))
    for _ in range(3):
        engine.train_step(ids, tgt)
    engine.save_checkpoint(path, save_optimizer=False)
    saved_w = model.emb.weight.detach().clone()

    torch.manual_seed(99)  # very different init
    with epl.replicate(device_count=1):
        model2 = TinyLM(sparse=True)
    engine2 = epl.Engine(model2, loss_fn=_ce, optimizer="adamw", lr=0.0)
    engine2.load_checkpoint(path, load_optimizer=False)
    assert torch.equal(model2.emb.weight.detach(), saved_w)
    engine2.train_step(ids, tgt)   # lr 0: params must NOT move
    return float((model2.emb.weight.detach() - saved_w).abs().max())


def test_sparse_master_refresh_on_partial_load(tmp_path):
    out = run_multiprocess(_worker_ckpt_no_opt, world=1,
                           args=(str(tmp_path),))
    assert out[0] == 0.0, out[0]
