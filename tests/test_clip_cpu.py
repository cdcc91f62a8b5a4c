"""Global grad-norm clipping (reference capability:
communication.clip_after_allreduce, epl/config.py:96-97): de-duplicated
global norm across DP/PP; clip folds into the fused optimizer scale."""

import torch
import torch.nn as nn

import easyparallellibrary_amd as epl
from tests.utils import run_multiprocess


def _build():
    torch.manual_seed(11)
    with epl.replicate(1):
        m = nn.Sequential(nn.Linear(8, 32), nn.Tanh(), nn.Linear(32, 4))
    return m


def test_global_grad_norm_matches_torch():
    epl.init()
    engine = epl.Engine(_build(), loss_fn=nn.MSELoss(), optimizer="adamw",
                        lr=1e-2)
    torch.manual_seed(12)
    x, y = torch.randn(16, 8), torch.randn(16, 4)
    engine.zero_grad()
    loss = engine.loss_fn(engine.model(x), y)
    loss.backward()
    engine.finish_grad_sync()
    want = torch.sqrt(sum((p.grad.float() ** 2).sum()
                          for p in engine.model.parameters()))
    got = engine._global_grad_norm()
    assert abs(got - float(want)) < 1e-5, (got, float(want))


def test_clip_changes_trajectory_only_when_binding():
    def run(max_norm):
        from easyparallellibrary_amd.env import Env
        from easyparallellibrary_amd.parallel import hooks
        hooks.remove_hooks()
        Env._instance = None
        epl.init(epl.Config({"optimizer.max_grad_norm": max_norm}))
        engine = epl.Engine(_build(), loss_fn=nn.MSELoss(),
                            optimizer="adamw", lr=1e-2)
        torch.manual_seed(12)
        x, y = torch.randn(16, 8), torch.randn(16, 4)
        return [float(engine.train_step(x, y)) for _ in range(3)]

    base = run(0.0)
    loose = run(1e9)     # never binds -> identical
    tight = run(1e-3)    # always binds -> different
    assert base == loose, (base, loose)
    assert base != tight


def _norm_worker(rank, world, pp):
    import easyparallellibrary_amd as epl
    del pp
    epl.init()
    torch.manual_seed(11)
    with epl.replicate(1):
        model = nn.Sequential(nn.Linear(8, 32), nn.Tanh(),
                              nn.Linear(32, 4))
    engine = epl.Engine(model, loss_fn=nn.MSELoss(), optimizer="adamw",
                        lr=1e-2)
    torch.manual_seed(12)
    x, y = torch.randn(16, 8), torch.randn(16, 4)
    engine.zero_grad()
    loss = engine.loss_fn(engine.model(x), y)
    loss.backward()
    engine.finish_grad_sync()
    return engine._global_grad_norm()


def test_global_norm_dedup_dp2():
    """DP2 (grads averaged identically on both ranks) reports the same
    global norm as the single-rank run."""
    single = _norm_worker(0, 1, False)
    # reset global state left by the in-process run
    from easyparallellibrary_amd.env import Env
    from easyparallellibrary_amd.parallel import hooks
    hooks.remove_hooks()
    Env._instance = None
    dp = run_multiprocess(_norm_worker, world=2, args=(False,))
    assert abs(dp[0] - dp[1]) < 1e-6
    assert abs(dp[0] - single) < 1e-4, (dp[0], single)


def test_parity_knobs():
    """Reference-parity config knobs: amp.debug_log, drop_last_files,
    gc end_taskgraph / check_gradients, clip_after_allreduce."""
    from easyparallellibrary_amd.utils.io_slicing import slice_files
    files = [str(i) for i in range(10)]
    parts = [slice_files(files, 4, r, drop_last=True) for r in range(4)]
    assert [len(p) for p in parts] == [2, 2, 2, 2]
    assert sum(parts, []) == files[:8]


def test_gc_check_gradients_logs_and_trains():
    from easyparallellibrary_amd.env import Env
    from easyparallellibrary_amd.parallel import hooks
    hooks.remove_hooks()
    Env._instance = None
    epl.init(epl.Config({"gradient_checkpoint.type": "auto",
                         "gradient_checkpoint.check_gradients": True}))
    torch.manual_seed(13)

    class Block(nn.Module):
        def __init__(self):
            super().__init__()
            self.a = nn.Linear(8, 8)

        def forward(self, x):
            return torch.tanh(self.a(x))

    with epl.replicate(1):
        model = nn.Sequential(Block(), Block(), Block(), nn.Linear(8, 2))
    engine = epl.Engine(model, loss_fn=nn.MSELoss(), optimizer="adamw",
                        lr=1e-2)
    assert engine._gc_wrapped == 3
    torch.manual_seed(14)
    x, y = torch.randn(8, 8), torch.randn(8, 2)
    losses = [float(engine.train_step(x, y)) for _ in range(2)]
    assert losses[1] < losses[0]


def _zero_norm_worker(rank, world, zero_level):
    import easyparallellibrary_amd as epl
    epl.init(epl.Config({"zero.level": zero_level}))
    torch.manual_seed(11)
    with epl.replicate(1):
        model = nn.Sequential(nn.Linear(8, 32), nn.Tanh(),
                              nn.Linear(32, 4))
    engine = epl.Engine(model, loss_fn=nn.MSELoss(), optimizer="adamw",
                        lr=1e-2)
    torch.manual_seed(12)
    x, y = torch.randn(16, 8), torch.randn(16, 4)
    engine.zero_grad()
    loss = engine.loss_fn(engine.model(x), y)
    loss.backward()
    engine.finish_grad_sync()
    return engine._global_grad_norm()


def test_global_norm_zero_v1_dedup():
    """Under ZeRO v1 only the owned shard is counted; the norm still
    equals the plain-DP value."""
    plain = run_multiprocess(_zero_norm_worker, world=2, args=("",))
    z1 = run_multiprocess(_zero_norm_worker, world=2, args=("v1",))
    assert abs(plain[0] - plain[1]) < 1e-6
    assert abs(z1[0] - z1[1]) < 1e-6
    assert abs(plain[0] - z1[0]) < 1e-5, (plain[0], z1[0])
