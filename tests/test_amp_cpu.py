"""AMP (O1) behavior: autocast wrapping, dynamic loss scale updates, and
overflow-step skipping (reference: tests/amp_test.py semantics)."""

import torch
import torch.nn as nn

import easyparallellibrary_amd as epl
from easyparallellibrary_amd.runtime.amp import DynamicLossScaler


def test_amp_o1_bf16_trains():
    epl.init(epl.Config({"amp.level": "O1", "amp.dtype": "bf16"}))
    torch.manual_seed(1)
    with epl.replicate(1):
        m = nn.Sequential(nn.Linear(8, 16), nn.Tanh(), nn.Linear(16, 2))
    engine = epl.Engine(m, loss_fn=nn.MSELoss(), lr=1e-2)
    assert engine.amp.enabled
    assert engine.amp.loss_scale == 1.0  # bf16 needs no scaling
    x = torch.randn(4, 8)
    y = torch.randn(4, 2)
    losses = [float(engine.train_step(x, y)) for _ in range(3)]
    assert losses[-1] < losses[0]


def test_amp_fp16_scaled_step_equivalence():
    """fp16 AMP with a fixed scale must track the unscaled trajectory
    (unscale is folded into the optimizer)."""
    def run(amp):
        from easyparallellibrary_amd.env import Env
        from easyparallellibrary_amd.parallel import hooks
        hooks.remove_hooks()
        Env._instance = None
        cfg = {"amp.level": "O1", "amp.dtype": "fp16",
               "amp.loss_scale": "1024"} if amp else {}
        epl.init(epl.Config(cfg))
        torch.manual_seed(2)
        with epl.replicate(1):
            m = nn.Sequential(nn.Linear(8, 16), nn.Tanh(), nn.Linear(16, 2))
        engine = epl.Engine(m, loss_fn=nn.MSELoss(), lr=1e-2)
        torch.manual_seed(3)
        x = torch.randn(16, 8)
        y = torch.randn(16, 2)
        return [float(engine.train_step(x, y)) for _ in range(3)]

    base = run(False)
    amp = run(True)
    # fp16 autocast on CPU still runs ops in fp32 where unsupported; the
    # trajectories must agree to fp16-ish tolerance
    for a, b in zip(base, amp):
        assert abs(a - b) < 5e-3, (base, amp)


def test_overflow_skips_step():
    epl.init(epl.Config({"amp.level": "O1", "amp.dtype": "fp16"}))
    torch.manual_seed(4)
    with epl.replicate(1):
        m = nn.Linear(4, 2)
    engine = epl.Engine(m, loss_fn=nn.MSELoss(), lr=1e-2)
    before = engine.flat_groups[0].param_arena.clone()
    x = torch.randn(2, 4)
    y = torch.randn(2, 2)
    engine.zero_grad()
    with engine.amp.autocast():
        loss = engine.loss_fn(engine.forward(x), y)
    engine.amp.scale_loss(loss).backward()
    engine.finish_grad_sync()
    # poison one grad with inf -> found_inf -> step skipped, scale backed off
    engine.flat_groups[0].grad_arena.view(-1)[0] = float("inf")
    s0 = engine.amp.loss_scale
    found = engine.amp.found_inf(engine.flat_groups)
    assert found
    engine.amp.post_step(found)
    assert engine.amp.loss_scale == s0 * 0.5
    assert torch.equal(before, engine.flat_groups[0].param_arena)


def test_dynamic_scaler_growth():
    s = DynamicLossScaler(init_scale=8, growth_interval=3)
    for _ in range(3):
        s.update(False)
    assert s.scale == 16


def _amp_pp_worker(rank, world):
    import easyparallellibrary_amd as epl
    from tests.utils import run_multiprocess  # noqa: F401 (import guard)
    epl.init(epl.Config({
        "amp.level": "O1", "amp.loss_scale": 128,
        "pipeline.num_micro_batch": 2,
    }))
    torch.manual_seed(21)
    with epl.replicate(1, name="stage_0"):
        s0 = nn.Sequential(nn.Linear(8, 16), nn.Tanh())
    with epl.replicate(1, name="stage_1"):
        s1 = nn.Linear(16, 4)

    class M(nn.Module):
        def __init__(self):
            super().__init__()
            self.s0, self.s1 = s0, s1

        def forward(self, x):
            return self.s1(self.s0(x))

    engine = epl.Engine(M(), loss_fn=nn.MSELoss(), optimizer="adamw",
                        lr=1e-2)
    torch.manual_seed(22)
    x, y = torch.randn(8, 8), torch.randn(8, 4)
    out = []
    for _ in range(3):
        loss = engine.train_step(x, y)
        out.append(None if loss is None else float(loss))
    return out


def test_amp_with_pipeline():
    """AMP loss scaling under PP2: the overflow verdict is allreduced so
    stages step in lockstep; training converges."""
    from tests.utils import run_multiprocess
    res = run_multiprocess(_amp_pp_worker, world=2)
    assert res[0][0] is None
    assert res[1][-1] < res[1][0], res[1]


def test_loss_scale_persists_across_checkpoint(tmp_path):
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.env import Env
    from easyparallellibrary_amd.parallel import hooks

    def build(seed):
        hooks.remove_hooks()
        Env._instance = None
        epl.init(epl.Config({"amp.level": "O1", "amp.dtype": "fp16",
                             "amp.loss_scale": "dynamic"}))
        torch.manual_seed(seed)
        with epl.replicate(1):
            m = nn.Sequential(nn.Linear(8, 8), nn.Tanh(), nn.Linear(8, 2))
        return epl.Engine(m, loss_fn=nn.MSELoss(), optimizer="adamw",
                          lr=1e-3)

    e1 = build(1)
    e1.amp.scaler.scale = 4096.0  # pretend overflows shrank it
    x, y = torch.randn(4, 8), torch.randn(4, 2)
    e1.train_step(x, y)
    scale_after = e1.amp.loss_scale
    e1.save_checkpoint(str(tmp_path))
    e2 = build(2)
    assert e2.amp.loss_scale != scale_after  # fresh default
    e2.load_checkpoint(str(tmp_path))
    assert e2.amp.loss_scale == scale_after
