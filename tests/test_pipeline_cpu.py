"""Pipeline runtime on CPU/gloo: PP2 (1F1B and GPipe) must match serial
training; micro-batch gradient accumulation must match big-batch."""

import torch
import torch.nn as nn

from tests.utils import run_multiprocess

HID = 16


def _build(world_stages):
    import easyparallellibrary_amd as epl
    torch.manual_seed(21)
    if world_stages == 1:
        with epl.replicate(device_count=1):
            model = nn.Sequential(
                nn.Linear(8, HID), nn.Tanh(), nn.Linear(HID, HID),
                nn.Tanh(), nn.Linear(HID, 4))
        return model
    with epl.replicate(device_count=1, name="stage_0"):
        s0 = nn.Sequential(nn.Linear(8, HID), nn.Tanh())
    with epl.replicate(device_count=1, name="stage_1"):
        s1 = nn.Sequential(nn.Linear(HID, HID), nn.Tanh(),
                           nn.Linear(HID, 4))

    class M(nn.Module):
        def __init__(self):
            super().__init__()
            self.s0, self.s1 = s0, s1

        def forward(self, x):
            return self.s1(self.s0(x))

    return M()


def _pipeline_worker(rank, world, schedule, nmb):
    import easyparallellibrary_amd as epl
    epl.init(epl.Config({
        "pipeline.num_micro_batch": nmb,
        "pipeline.strategy": schedule,
    }))
    model = _build(world_stages=world)
    engine = epl.Engine(model, loss_fn=nn.MSELoss(), optimizer="adamw",
                        lr=1e-2)
    torch.manual_seed(33)
    x = torch.randn(8, 8)
    y = torch.randn(8, 4)
    losses = []
    for _ in range(3):
        loss = engine.train_step(x, y)
        losses.append(None if loss is None else float(loss))
    return losses


def _serial_worker(rank, world, nmb):
    import easyparallellibrary_amd as epl
    epl.init(epl.Config({"pipeline.num_micro_batch": nmb}))
    model = _build(world_stages=1)
    engine = epl.Engine(model, loss_fn=nn.MSELoss(), optimizer="adamw",
                        lr=1e-2)
    torch.manual_seed(33)
    x = torch.randn(8, 8)
    y = torch.randn(8, 4)
    return [float(engine.train_step(x, y)) for _ in range(3)]


def test_pp2_1f1b_matches_serial():
    serial = run_multiprocess(_serial_worker, world=1, args=(4,))[0]
    pp = run_multiprocess(_pipeline_worker, world=2,
                          args=("prefer_backward", 4))
    # loss lives on the last stage (rank 1)
    assert pp[0][0] is None
    assert all(abs(a - b) < 1e-5 for a, b in zip(serial, pp[1]))


def test_pp2_gpipe_matches_serial():
    serial = run_multiprocess(_serial_worker, world=1, args=(4,))[0]
    pp = run_multiprocess(_pipeline_worker, world=2,
                          args=("prefer_forward", 4))
    assert all(abs(a - b) < 1e-5 for a, b in zip(serial, pp[1]))


def test_micro_batch_accumulation_matches_big_batch():
    big = run_multiprocess(_serial_worker, world=1, args=(1,))[0]
    acc = run_multiprocess(_serial_worker, world=1, args=(4,))[0]
    assert all(abs(a - b) < 1e-5 for a, b in zip(big, acc))


def test_pp2_dp2_runs():
    """2 stages x 2 replicas on 4 ranks: full hybrid."""
    pp = run_multiprocess(_pipeline_worker, world=4,
                          args=("prefer_backward", 4))
    # last-stage ranks (1 and 3) report identical losses (same data/seed)
    assert pp[1] == pp[3]
    assert pp[1][-1] < pp[1][0]


def test_pp2_fewer_microbatches_than_stages():
    """M=1 < S=2: warmup clamps; still matches serial."""
    serial = run_multiprocess(_serial_worker, world=1, args=(1,))[0]
    pp = run_multiprocess(_pipeline_worker, world=2,
                          args=("prefer_backward", 1))
    assert all(abs(a - b) < 1e-5 for a, b in zip(serial, pp[1]))


def test_pp2_odd_microbatches():
    serial = run_multiprocess(_serial_worker, world=1, args=(3,))[0]
    pp = run_multiprocess(_pipeline_worker, world=2,
                          args=("prefer_backward", 3))
    assert all(abs(a - b) < 1e-5 for a, b in zip(serial, pp[1]))


def _pipeline_gc_worker(rank, world, gc_type):
    import easyparallellibrary_amd as epl
    epl.init(epl.Config({
        "pipeline.num_micro_batch": 4,
        "gradient_checkpoint.type": gc_type,
    }))
    model = _build(world_stages=world)
    engine = epl.Engine(model, loss_fn=nn.MSELoss(), optimizer="adamw",
                        lr=1e-2)
    torch.manual_seed(33)
    x = torch.randn(8, 8)
    y = torch.randn(8, 4)
    out = []
    for _ in range(3):
        loss = engine.train_step(x, y)
        out.append(None if loss is None else float(loss))
    return out


def test_pp2_with_gradient_checkpoint():
    """Recompute inside pipeline stages: same losses as without GC."""
    plain = run_multiprocess(_pipeline_worker, world=2,
                             args=("prefer_backward", 4))
    gc = run_multiprocess(_pipeline_gc_worker, world=2, args=("auto",))
    assert all(abs(a - b) < 1e-5 for a, b in zip(plain[1], gc[1])), (
        plain[1], gc[1])


def test_pp2_prefer_backward_optimizer_matches_serial():
    """PBO is a REAL schedule now: buckets apply eagerly during the
    final backward via reducer callbacks (reference scheduler.py:87-116)
    and the trajectory must still match serial exactly."""
    serial = run_multiprocess(_serial_worker, world=1, args=(4,))[0]
    pp = run_multiprocess(_pipeline_worker, world=2,
                          args=("prefer_backward_optimizer", 4))
    assert pp[0][0] is None
    assert all(abs(a - b) < 1e-5 for a, b in zip(serial, pp[1])), (
        serial, pp[1])


def _pbo_eager_active_worker(rank, world):
    import easyparallellibrary_amd as epl
    epl.init(epl.Config({
        "pipeline.num_micro_batch": 4,
        "pipeline.strategy": "prefer_backward_optimizer",
    }))
    model = _build(world_stages=world)
    engine = epl.Engine(model, loss_fn=nn.MSELoss(), optimizer="adamw",
                        lr=1e-2)
    torch.manual_seed(33)
    engine.train_step(torch.randn(8, 8), torch.randn(8, 4))
    return engine._pbo_eager, engine.optimizer.step_count


def test_pbo_eager_path_is_active():
    res = run_multiprocess(_pbo_eager_active_worker, world=2)
    for eager, steps in res:
        assert eager is True
        assert steps == 1


def _eval_pp_worker(rank, world):
    import easyparallellibrary_amd as epl
    epl.init(epl.Config({"pipeline.num_micro_batch": 4}))
    model = _build(world_stages=world)
    engine = epl.Engine(model, loss_fn=nn.MSELoss(), optimizer="adamw",
                        lr=1e-2)
    torch.manual_seed(33)
    x = torch.randn(8, 8)
    out = engine.eval_step(x)
    # a second eval with a DIFFERENT batch size (shape handshake is
    # per-call, not cached from training)
    out2 = engine.eval_step(torch.randn(4, 8))
    engine.close()
    return (None if out is None else out.clone(),
            None if out2 is None else out2.shape)


def _eval_serial_worker(rank, world):
    import easyparallellibrary_amd as epl
    epl.init()
    model = _build(world_stages=1)
    engine = epl.Engine(model, loss_fn=nn.MSELoss())
    torch.manual_seed(33)
    x = torch.randn(8, 8)
    return engine.eval_step(x).clone()


def test_pipelined_eval_matches_serial():
    """run_eval: forward-only pipelined evaluation — last stage output
    must equal the serial model's eval output; works with eval batch
    shapes different from training (fresh shape handshake)."""
    serial = run_multiprocess(_eval_serial_worker, world=1)[0]
    pp = run_multiprocess(_eval_pp_worker, world=2)
    assert pp[0][0] is None             # stage 0 yields no output
    assert torch.allclose(pp[1][0], serial, atol=1e-6)
    assert tuple(pp[1][1]) == (4, 4)    # different eval batch worked


def _close_worker(rank, world):
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.comm import backend
    epl.init()
    with epl.replicate(device_count=1):
        model = nn.Linear(4, 2)
    engine = epl.Engine(model, loss_fn=nn.MSELoss())
    before = len(backend._REGISTRY)
    engine.close()
    after = len(backend._REGISTRY)
    return before, after


def test_engine_close_shrinks_registry():
    before, after = run_multiprocess(_close_worker, world=1)[0]
    assert before > 0 and after == 0, (before, after)


def _gc_verify_pp_worker(rank, world):
    """check_gradients now runs under pipeline too: the schedule reruns
    with recompute off and the local stage's gradients must match."""
    import easyparallellibrary_amd as epl
    epl.init(epl.Config({
        "pipeline.num_micro_batch": 4,
        "gradient_checkpoint.type": "auto",
        "gradient_checkpoint.check_gradients": True,
    }))
    model = _build(world_stages=world)
    engine = epl.Engine(model, loss_fn=nn.MSELoss(), optimizer="adamw",
                        lr=1e-2)
    torch.manual_seed(33)
    x = torch.randn(8, 8)
    y = torch.randn(8, 4)
    losses = [engine.train_step(x, y) for _ in range(2)]
    return [None if l is None else float(l) for l in losses]


def test_gc_check_gradients_under_pipeline():
    res = run_multiprocess(_gc_verify_pp_worker, world=2)
    assert res[1][0] is not None and res[1][1] is not None
