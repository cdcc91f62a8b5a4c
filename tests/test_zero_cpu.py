"""ZeRO v0/v1 on CPU/gloo: sharded state + (v1) reduce-to-owner gradients
must reproduce plain DP training exactly (reference: tests/zero_test.py)."""

import torch
import torch.nn as nn

from tests.utils import run_multiprocess


def _worker(rank, world, zero_level):
    import easyparallellibrary_amd as epl
    epl.init(epl.Config({"zero.level": zero_level}))
    torch.manual_seed(17)
    with epl.replicate(device_count=1):
        model = nn.Sequential(nn.Linear(16, 64), nn.Tanh(),
                              nn.Linear(64, 64), nn.Tanh(),
                              nn.Linear(64, 4))
    engine = epl.Engine(model, loss_fn=nn.MSELoss(), optimizer="adamw",
                        lr=1e-2)
    torch.manual_seed(23)
    x = torch.randn(8, 16)
    y = torch.randn(8, 4)
    losses = [float(engine.train_step(x, y)) for _ in range(4)]
    return losses, engine.flat_groups[0].param_arena.clone()


def test_zero_v0_matches_dp():
    base = run_multiprocess(_worker, world=2, args=("",))
    z0 = run_multiprocess(_worker, world=2, args=("v0",))
    assert all(abs(a - b) < 1e-5 for a, b in zip(base[0][0], z0[0][0]))
    n = min(base[0][1].numel(), z0[0][1].numel())
    assert torch.allclose(base[0][1][:n], z0[0][1][:n], atol=1e-6)
    assert torch.allclose(z0[0][1], z0[1][1], atol=0)  # ranks in sync


def test_zero_v1_matches_dp():
    base = run_multiprocess(_worker, world=2, args=("",))
    z1 = run_multiprocess(_worker, world=2, args=("v1",))
    assert all(abs(a - b) < 1e-5 for a, b in zip(base[0][0], z1[0][0]))
    n = min(base[0][1].numel(), z1[0][1].numel())
    assert torch.allclose(base[0][1][:n], z1[0][1][:n], atol=1e-6)
    assert torch.allclose(z1[0][1], z1[1][1], atol=0)
