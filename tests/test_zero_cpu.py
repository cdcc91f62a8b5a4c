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


def _pp_zero_worker(rank, world, zero_level):
    """PP2 x DP(world/2) with ZeRO sharding across each stage's DP group."""
    import easyparallellibrary_amd as epl
    epl.init(epl.Config({"zero.level": zero_level,
                         "pipeline.num_micro_batch": 2}))
    torch.manual_seed(40)
    with epl.replicate(device_count=1, name="stage_0"):
        s0 = nn.Sequential(nn.Linear(16, 64), nn.Tanh())
    with epl.replicate(device_count=1, name="stage_1"):
        s1 = nn.Sequential(nn.Linear(64, 64), nn.Tanh(), nn.Linear(64, 4))

    class M(nn.Module):
        def __init__(self):
            super().__init__()
            self.s0, self.s1 = s0, s1

        def forward(self, x):
            return self.s1(self.s0(x))

    engine = epl.Engine(M(), loss_fn=nn.MSELoss(), optimizer="adamw",
                        lr=1e-2)
    torch.manual_seed(41)
    x = torch.randn(8, 16)
    y = torch.randn(8, 4)
    out = []
    for _ in range(4):
        loss = engine.train_step(x, y)
        out.append(None if loss is None else float(loss))
    return out


def test_zero_v1_with_pipeline():
    """ZeRO shards across each stage's 2-replica DP group under PP2."""
    base = run_multiprocess(_pp_zero_worker, world=4, args=("",),
                            timeout=300)
    z1 = run_multiprocess(_pp_zero_worker, world=4, args=("v1",),
                          timeout=300)
    # last-stage ranks: 1 and 3
    assert base[1] == base[3] and z1[1] == z1[3]
    assert all(abs(a - b) < 1e-5 for a, b in zip(base[1], z1[1])), (
        base[1], z1[1])


def _comp_worker(rank, world, zero_level, comp):
    import easyparallellibrary_amd as epl
    epl.init(epl.Config({"zero.level": zero_level,
                         "communication.compression": comp}))
    torch.manual_seed(17)
    with epl.replicate(device_count=1):
        model = nn.Sequential(nn.Linear(16, 64), nn.Tanh(),
                              nn.Linear(64, 4))
    engine = epl.Engine(model, loss_fn=nn.MSELoss(), optimizer="adamw",
                        lr=1e-2)
    torch.manual_seed(23)
    x = torch.randn(8, 16)
    y = torch.randn(8, 4)
    return [float(engine.train_step(x, y)) for _ in range(4)]


def test_zero_v1_with_wire_compression():
    """bf16-compressed reduce-to-owner buckets track the fp32 wire to
    bf16 rounding error."""
    base = run_multiprocess(_comp_worker, world=2, args=("v1", ""))
    comp = run_multiprocess(_comp_worker, world=2, args=("v1", "bf16"))
    assert all(abs(a - b) < 1e-3 for a, b in zip(base[0], comp[0])), (
        base[0], comp[0])
