"""Engine odds and ends: eval, scheduler variants, GA+pipeline combos,
non-overlap reduction path."""

import torch
import torch.nn as nn

from tests.utils import run_multiprocess


def test_eval_step():
    import easyparallellibrary_amd as epl
    epl.init()
    torch.manual_seed(1)
    with epl.replicate(1):
        m = nn.Linear(4, 2)
    engine = epl.Engine(m, loss_fn=nn.MSELoss())
    out = engine.eval_step(torch.randn(3, 4))
    assert out.shape == (3, 2)
    assert not out.requires_grad


def _pbo_worker(rank, world):
    import easyparallellibrary_amd as epl
    epl.init(epl.Config({
        "pipeline.num_micro_batch": 4,
        "pipeline.strategy": "prefer_backward_optimizer",
    }))
    torch.manual_seed(21)
    with epl.replicate(device_count=1, name="stage_0"):
        s0 = nn.Sequential(nn.Linear(8, 16), nn.Tanh())
    with epl.replicate(device_count=1, name="stage_1"):
        s1 = nn.Linear(16, 4)

    class M(nn.Module):
        def __init__(self):
            super().__init__()
            self.s0, self.s1 = s0, s1

        def forward(self, x):
            return self.s1(self.s0(x))

    engine = epl.Engine(M(), loss_fn=nn.MSELoss(), lr=1e-2)
    torch.manual_seed(22)
    x = torch.randn(8, 8)
    y = torch.randn(8, 4)
    out = [engine.train_step(x, y) for _ in range(3)]
    return [None if o is None else float(o) for o in out]


def test_prefer_backward_optimizer_schedule():
    res = run_multiprocess(_pbo_worker, world=2)
    losses = res[1]
    assert losses[-1] < losses[0]


def _no_overlap_worker(rank, world):
    import easyparallellibrary_amd as epl
    epl.init(epl.Config({"communication.overlap_grad_reduce": False}))
    torch.manual_seed(31)
    with epl.replicate(1):
        m = nn.Sequential(nn.Linear(8, 8), nn.Tanh(), nn.Linear(8, 2))
    engine = epl.Engine(m, loss_fn=nn.MSELoss(), lr=1e-2)
    torch.manual_seed(32)
    x = torch.randn(4, 8)
    y = torch.randn(4, 2)
    return [float(engine.train_step(x, y)) for _ in range(3)]


def test_no_overlap_matches_overlap():
    off = run_multiprocess(_no_overlap_worker, world=2)
    assert off[0] == off[1]
    assert off[0][-1] < off[0][0]


def _bucket_sizes_worker(rank, world):
    """Tiny bucket_bytes -> many buckets; trajectory must be unchanged."""
    import easyparallellibrary_amd as epl
    epl.init(epl.Config({"communication.bucket_bytes": 256,
                         "communication.num_communicators": 3}))
    torch.manual_seed(41)
    with epl.replicate(1):
        m = nn.Sequential(nn.Linear(16, 64), nn.Tanh(), nn.Linear(64, 4))
    engine = epl.Engine(m, loss_fn=nn.MSELoss(), lr=1e-2)
    assert len(engine.reducers[0].buckets) > 2
    torch.manual_seed(42)
    x = torch.randn(4, 16)
    y = torch.randn(4, 4)
    return [float(engine.train_step(x, y)) for _ in range(3)]


def test_many_small_buckets():
    res = run_multiprocess(_bucket_sizes_worker, world=2)
    assert res[0] == res[1]
    assert res[0][-1] < res[0][0]


def _compress_worker(rank, world):
    import easyparallellibrary_amd as epl
    epl.init(epl.Config({"communication.compression": "bf16"}))
    torch.manual_seed(51)
    with epl.replicate(1):
        m = nn.Sequential(nn.Linear(8, 16), nn.Tanh(), nn.Linear(16, 2))
    engine = epl.Engine(m, loss_fn=nn.MSELoss(), lr=1e-2)
    torch.manual_seed(52)
    x = torch.randn(4, 8)
    y = torch.randn(4, 2)
    return [float(engine.train_step(x, y)) for _ in range(3)]


def test_grad_compression():
    res = run_multiprocess(_compress_worker, world=2)
    assert res[0] == res[1]
    assert res[0][-1] < res[0][0]


def test_set_lr():
    import easyparallellibrary_amd as epl
    epl.init()
    torch.manual_seed(1)
    with epl.replicate(1):
        model = nn.Linear(4, 2)
    engine = epl.Engine(model, loss_fn=nn.MSELoss(), optimizer="adamw",
                        lr=1e-2)
    assert engine.lr == 1e-2
    x, y = torch.randn(4, 4), torch.randn(4, 2)
    engine.train_step(x, y)
    w0 = model.weight.detach().clone()
    engine.set_lr(0.0)
    engine.train_step(x, y)
    # lr=0 scales the whole update (incl. decoupled weight decay) to
    # zero -> parameters must be bit-identical
    assert torch.equal(model.weight.detach(), w0)


def test_kernel_fused_off_still_trains():
    """kernel.fused=False forces the eager op paths everywhere."""
    import easyparallellibrary_amd as epl
    epl.init(epl.Config({"kernel.fused": False}))
    torch.manual_seed(3)
    with epl.replicate(1):
        model = nn.Sequential(nn.Linear(8, 16), nn.Tanh(),
                              nn.Linear(16, 2))
    engine = epl.Engine(model, loss_fn=nn.MSELoss(), optimizer="adamw",
                        lr=1e-2)
    x, y = torch.randn(8, 8), torch.randn(8, 2)
    losses = [float(engine.train_step(x, y)) for _ in range(3)]
    assert losses[-1] < losses[0]


def test_manual_accumulation_matches_big_batch():
    import easyparallellibrary_amd as epl

    def build_engine():
        from easyparallellibrary_amd.env import Env
        from easyparallellibrary_amd.parallel import hooks
        hooks.remove_hooks()
        Env._instance = None
        epl.init()
        torch.manual_seed(5)
        with epl.replicate(1):
            m = nn.Sequential(nn.Linear(8, 16), nn.Tanh(),
                              nn.Linear(16, 2))
        return epl.Engine(m, loss_fn=nn.MSELoss(), optimizer="adamw",
                          lr=1e-2)

    torch.manual_seed(6)
    x = torch.randn(8, 8)
    y = torch.randn(8, 2)

    e1 = build_engine()
    big = [float(e1.train_step(x, y)) for _ in range(3)]

    e2 = build_engine()
    man = []
    for _ in range(3):
        l0 = e2.train_step(x[:4], y[:4], accumulate=True)
        l1 = e2.train_step(x[4:], y[4:])
        man.append((float(l0) + float(l1)) / 2)
    assert all(abs(a - b) < 1e-6 for a, b in zip(big, man)), (big, man)


def test_random_architectures_train():
    """Robustness sweep: a handful of randomized architectures step
    twice through the engine without shape/edge assumptions breaking."""
    import random
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.env import Env
    from easyparallellibrary_amd.parallel import hooks
    rng = random.Random(1234)
    for trial in range(5):
        hooks.remove_hooks()
        Env._instance = None
        epl.init()
        torch.manual_seed(trial)
        depth = rng.randint(1, 4)
        dims = [rng.choice([3, 8, 17, 32])]
        layers = []
        for _ in range(depth):
            nxt = rng.choice([5, 16, 31])
            layers += [nn.Linear(dims[-1], nxt), nn.Tanh()]
            dims.append(nxt)
        layers.append(nn.Linear(dims[-1], 4))
        with epl.replicate(1):
            model = nn.Sequential(*layers)
        engine = epl.Engine(model, loss_fn=nn.MSELoss(),
                            optimizer=rng.choice(["adamw", "lamb"]),
                            lr=1e-3)
        b = rng.choice([1, 3, 8])
        x, y = torch.randn(b, dims[0]), torch.randn(b, 4)
        for _ in range(2):
            loss = engine.train_step(x, y)
        assert torch.isfinite(loss), (trial, loss)


def test_eval_step_uses_eval_mode():
    """eval_step must not update BatchNorm running stats or apply
    dropout, and must restore training mode afterwards."""
    import easyparallellibrary_amd as epl
    epl.init()
    torch.manual_seed(4)
    with epl.replicate(1):
        model = nn.Sequential(nn.Linear(8, 8), nn.BatchNorm1d(8),
                              nn.Dropout(0.5), nn.Linear(8, 2))
    engine = epl.Engine(model, loss_fn=nn.MSELoss(), optimizer="adamw",
                        lr=1e-3)
    x = torch.randn(16, 8)
    engine.train_step(x, torch.randn(16, 2))
    rm = model[1].running_mean.clone()
    y1 = engine.eval_step(x)
    y2 = engine.eval_step(x)
    assert torch.equal(y1, y2)                     # dropout off
    assert torch.equal(model[1].running_mean, rm)  # stats frozen
    assert model.training                          # mode restored


def test_bf16_training_on_cpu():
    """Pure-bf16 engine on CPU (eager fallbacks + fp32 master)."""
    import easyparallellibrary_amd as epl
    epl.init()
    torch.manual_seed(0)
    with epl.replicate(1):
        model = nn.Sequential(nn.Linear(8, 16), nn.Tanh(),
                              nn.Linear(16, 2))
    engine = epl.Engine(model, loss_fn=nn.MSELoss(), optimizer="adamw",
                        lr=1e-2, dtype=torch.bfloat16)
    x = torch.randn(8, 8).bfloat16()
    y = torch.randn(8, 2).bfloat16()
    losses = [float(engine.train_step(x, y)) for _ in range(3)]
    assert losses[-1] < losses[0]
    assert model[0].weight.dtype == torch.bfloat16


def test_no_memory_growth_over_steps():
    """200 steps leave RSS flat (hooks/pending-tensor bookkeeping must
    not accumulate)."""
    import gc
    import psutil
    import easyparallellibrary_amd as epl
    epl.init()
    torch.manual_seed(0)
    with epl.replicate(1):
        model = nn.Sequential(nn.Linear(64, 256), nn.Tanh(),
                              nn.Linear(256, 64))
    engine = epl.Engine(model, loss_fn=nn.MSELoss(), optimizer="adamw",
                        lr=1e-3)
    x, y = torch.randn(32, 64), torch.randn(32, 64)
    proc = psutil.Process()
    for _ in range(20):
        engine.train_step(x, y)
    gc.collect()
    rss0 = proc.memory_info().rss
    for _ in range(200):
        engine.train_step(x, y)
    gc.collect()
    growth = proc.memory_info().rss - rss0
    assert growth < 64 * 1024 * 1024, "RSS grew {} bytes".format(growth)


def _summaries_worker(rank, world):
    import easyparallellibrary_amd as epl
    epl.init()
    torch.manual_seed(31)
    with epl.replicate(1):
        m = nn.Linear(4, 2)
    engine = epl.Engine(m, loss_fn=nn.MSELoss())
    # once-registered callable (plays the reference's live summary tensor)
    epl.add_to_collection(("lr", lambda: engine.lr),
                          epl.GraphKeys.SUMMARIES)

    class FakeWriter:
        def __init__(self):
            self.rows = []

        def add_scalar(self, name, value, step):
            self.rows.append((name, value, step))

    w = FakeWriter()
    # per-rank value 10*rank: mean across 2 ranks = 5.0
    merged = engine.write_summaries(w, step=3,
                                    scalars={"loss": 10.0 * rank})
    return merged, w.rows


def test_write_summaries_merges_across_ranks():
    out = run_multiprocess(_summaries_worker, world=2)
    for rank in (0, 1):
        merged, rows = out[rank]
        assert abs(merged["loss"] - 5.0) < 1e-6, merged
        assert abs(merged["lr"] - 1e-3) < 1e-8
    # only rank 0 writes
    names = [(n, s) for n, _, s in out[0][1]]
    assert names == [("loss", 3), ("lr", 3)]
    assert abs(out[0][1][0][1] - 5.0) < 1e-6
    assert out[1][1] == []


def _signal_worker(rank, world):
    import time
    import easyparallellibrary_amd as epl
    epl.init()
    with epl.replicate(1):
        m = nn.Linear(4, 2)
    engine = epl.Engine(m, loss_fn=nn.MSELoss())
    if rank == 0:
        time.sleep(0.3)  # "chief evaluates"; others block on the signal
        return engine.broadcast_signal(7.5)
    return engine.broadcast_signal(-1.0)  # non-root value is ignored


def test_broadcast_signal_eval_barrier():
    out = run_multiprocess(_signal_worker, world=2)
    assert out == [7.5, 7.5]


def _drift_worker(rank, world):
    import torch
    import torch.nn as nn
    import easyparallellibrary_amd as epl
    epl.init()
    torch.manual_seed(12)
    with epl.replicate(device_count=1):
        model = nn.Linear(4, 2)
    engine = epl.Engine(model, loss_fn=nn.MSELoss())
    clean = engine.check_param_consistency()
    if rank == 1:   # inject drift on one replica
        engine.flat_groups[0].param_arena[0] += 0.5
    drifted = engine.check_param_consistency()
    engine.resync_params()
    fixed = engine.check_param_consistency()
    return clean, drifted, fixed


def test_param_consistency_check_and_resync():
    from tests.utils import run_multiprocess
    res = run_multiprocess(_drift_worker, world=2)
    for clean, drifted, fixed in res:
        assert clean == 0.0
        assert drifted > 0.1
        assert fixed == 0.0
