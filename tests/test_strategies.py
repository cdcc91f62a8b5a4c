"""Strategy annotation / context rules (reference: tests/strategy_test.py)."""

import pytest
import torch.nn as nn

import easyparallellibrary_amd as epl
from easyparallellibrary_amd.env import Env
from easyparallellibrary_amd.ir.plan import Plan
from easyparallellibrary_amd.parallel import hooks


def test_nesting_rules():
    epl.init()
    ctx = Env.get().strategy_context
    with epl.replicate(1, name="a"):
        with pytest.raises(ValueError):
            with epl.replicate(1, name="b"):
                pass
        with pytest.raises(ValueError):
            with epl.split(2, name="s"):
                pass
    with epl.split(2, name="s2"):
        with pytest.raises(ValueError):
            with epl.replicate(1, name="c"):
                pass
    assert ctx.num_taskgraphs >= 2


def test_same_name_same_taskgraph():
    epl.init()
    ctx = Env.get().strategy_context
    with epl.replicate(1, name="stage_0"):
        pass
    with epl.replicate(1, name="stage_0"):
        pass
    assert ctx.num_taskgraphs == 1


def test_module_tagging_and_plan():
    epl.init()
    with epl.replicate(1, name="stage_0"):
        a = nn.Linear(4, 4)
    with epl.replicate(1, name="stage_1"):
        b = nn.Linear(4, 4)

    class M(nn.Module):
        def __init__(self):
            super().__init__()
            self.a, self.b = a, b

    m = M()
    assert hooks.strategy_of(a).name == "stage_0"
    plan = Plan.build(m, Env.get().strategy_context)
    assert len(plan.taskgraphs) == 2
    assert plan.taskgraphs[0].module_names == ["a"]
    assert plan.taskgraphs[1].module_names == ["b"]
    assert plan.num_stages == 2


def test_default_strategy_collects_untagged():
    epl.init()
    epl.set_default_strategy(epl.replicate(1))
    m = nn.Sequential(nn.Linear(3, 3), nn.Linear(3, 3))
    plan = Plan.build(m, Env.get().strategy_context)
    assert len(plan.taskgraphs) == 1
    assert sum(1 for _ in plan.taskgraphs[0].parameters()) == 4


def test_split_plan_device_counts():
    epl.init()
    with epl.replicate(8, name="r"):
        backbone = nn.Linear(4, 4)
    with epl.split(8, name="s"):
        head = nn.Linear(4, 10)

    class M(nn.Module):
        def __init__(self):
            super().__init__()
            self.backbone, self.head = backbone, head

    plan = Plan.build(M(), Env.get().strategy_context)
    assert plan.effective_device_counts(False) == [8, 8]
    assert plan.effective_device_counts(True) == [8, 0]


def test_auto_parallel_mode():
    """auto.auto_parallel: untagged models train (whole model replicate);
    explicit scopes are rejected (reference parallel_strategy.py:61)."""
    import torch
    import pytest
    epl.init(epl.Config({"auto.auto_parallel": True}))
    with pytest.raises(ValueError):
        with epl.replicate(1):
            pass
    torch.manual_seed(2)
    model = nn.Sequential(nn.Linear(8, 8), nn.Tanh(), nn.Linear(8, 2))
    engine = epl.Engine(model, loss_fn=nn.MSELoss(), optimizer="adamw",
                        lr=1e-2)
    x, y = torch.randn(4, 8), torch.randn(4, 2)
    losses = [float(engine.train_step(x, y)) for _ in range(3)]
    assert losses[-1] < losses[0]
