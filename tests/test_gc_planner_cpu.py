"""Gradient checkpointing selection + auto-stage planner + profiler units."""

import torch
import torch.nn as nn

from tests.utils import run_multiprocess


def test_partition_balance():
    from easyparallellibrary_amd.parallel.partitioner import (
        find_repeated_blocks, partition_balance)
    parts = partition_balance([1, 1, 1, 1], 2)
    assert parts == [[0, 1], [2, 3]]
    parts = partition_balance([10, 1, 1, 1, 1], 2)
    assert parts == [[0], [1, 2, 3, 4]]
    assert find_repeated_blocks(["a", "b", "b", "b"], key=lambda x: x) == "b"
    assert find_repeated_blocks(["a", "b"], key=lambda x: x) is None


def test_gc_auto_selects_repeated_blocks():
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.runtime.gc import (
        CheckpointWrapper, apply_gradient_checkpointing)
    epl.init()

    class Block(nn.Module):
        def __init__(self):
            super().__init__()
            self.fc = nn.Linear(8, 8)

        def forward(self, x):
            return torch.tanh(self.fc(x))

    model = nn.Sequential(nn.Linear(8, 8), *[Block() for _ in range(4)],
                          nn.Linear(8, 2))
    n = apply_gradient_checkpointing(model, mode="auto")
    assert n == 4
    wrapped = [m for m in model.modules()
               if isinstance(m, CheckpointWrapper)]
    assert len(wrapped) == 4
    # gradients flow identically
    x = torch.randn(4, 8)
    y = model(x).sum()
    y.backward()
    assert all(p.grad is not None for p in model.parameters())


def test_gc_collection_mode():
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.runtime.gc import (
        annotate_checkpoint, apply_gradient_checkpointing)
    epl.init()
    inner = nn.Linear(4, 4)
    annotate_checkpoint(inner)
    model = nn.Sequential(inner, nn.Linear(4, 2))
    assert apply_gradient_checkpointing(model, mode="collection") == 1


def _auto_stage_worker(rank, world):
    import easyparallellibrary_amd as epl
    epl.init(epl.Config({"pipeline.num_stages": 2,
                         "pipeline.num_micro_batch": 4}))
    epl.set_default_strategy(epl.replicate(1))
    torch.manual_seed(50)
    model = nn.Sequential(nn.Linear(8, 32), nn.Tanh(), nn.Linear(32, 32),
                          nn.Tanh(), nn.Linear(32, 4))
    engine = epl.Engine(model, loss_fn=nn.MSELoss(), optimizer="adamw",
                        lr=1e-2)
    torch.manual_seed(51)
    x = torch.randn(8, 8)
    y = torch.randn(8, 4)
    losses = [engine.train_step(x, y) for _ in range(3)]
    losses = [None if l is None else float(l) for l in losses]
    return engine.num_stages, losses


def test_auto_stage_pipeline():
    res = run_multiprocess(_auto_stage_worker, world=2)
    assert res[0][0] == 2 and res[1][0] == 2
    assert res[0][1][0] is None           # stage-0 rank: no loss
    l = res[1][1]
    assert l[-1] < l[0]


def test_gc_training_matches_no_gc():
    import easyparallellibrary_amd as epl

    def run(gc):
        from easyparallellibrary_amd.env import Env
        from easyparallellibrary_amd.parallel import hooks
        hooks.remove_hooks()
        Env._instance = None
        cfg = {"gradient_checkpoint.type": "auto"} if gc else {}
        epl.init(epl.Config(cfg))
        torch.manual_seed(60)

        class Block(nn.Module):
            def __init__(self):
                super().__init__()
                self.fc = nn.Linear(8, 8)

            def forward(self, x):
                return torch.tanh(self.fc(x))

        with epl.replicate(1):
            model = nn.Sequential(nn.Linear(8, 8),
                                  *[Block() for _ in range(3)],
                                  nn.Linear(8, 2))
        engine = epl.Engine(model, loss_fn=nn.MSELoss(), optimizer="adamw",
                            lr=1e-2)
        torch.manual_seed(61)
        x = torch.randn(4, 8)
        y = torch.randn(4, 2)
        return [float(engine.train_step(x, y)) for _ in range(3)]

    base = run(False)
    gc = run(True)
    assert all(abs(a - b) < 1e-6 for a, b in zip(base, gc))


def test_cost_model_and_profiler():
    from easyparallellibrary_amd.profiler import (FlopsProfiler,
                                                  MemoryProfiler,
                                                  profile_memory)
    model = nn.Sequential(nn.Linear(16, 32), nn.Linear(32, 8))
    fp = FlopsProfiler(model, seq_len=10, batch=1)
    assert fp.total() == 2 * 16 * 32 * 10 + 2 * 32 * 8 * 10
    mem = profile_memory(model)
    assert all(v > 0 for v in mem.values())
    mp = MemoryProfiler()
    mp.after_step(1)
    assert len(mp.records) == 1


def test_partition_balance_randomized():
    """Contiguity/coverage + optimality bound: DP max-chunk cost is never
    above the trivial upper bound (total) and never below total/k."""
    import random
    from easyparallellibrary_amd.parallel.partitioner import (
        partition_balance)
    rng = random.Random(12)
    for _ in range(100):
        n = rng.randint(1, 40)
        k = rng.randint(1, 8)
        w = [rng.uniform(0.1, 10.0) for _ in range(n)]
        parts = partition_balance(w, k)
        assert len(parts) == min(k, n)
        flat = sum(parts, [])
        assert flat == list(range(n))          # contiguous, in order
        assert all(parts)                       # no empty chunk
        costs = [sum(w[i] for i in p) for p in parts]
        assert max(costs) >= sum(w) / len(parts) - 1e-9
        # DP optimality sanity: no single element exceeds... the max chunk
        # must be at least the largest single weight
        assert max(costs) >= max(w) - 1e-9


def test_gc_auto_never_wraps_fused_elementwise():
    """Regression: with few repeated blocks, auto-GC must not fall back
    to wrapping FusedLayerNorm/FusedBiasGelu (their duck-typed methods —
    forward_with_sum — break under a wrapper)."""
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.models import gpt2
    from easyparallellibrary_amd.runtime.gc import (
        select_checkpoint_modules)
    epl.init()
    model = gpt2.build_gpt2("gpt2-tiny", vocab_size=128, max_pos=32)
    hits = select_checkpoint_modules(model, mode="auto")
    from easyparallellibrary_amd.ops.layer_norm import FusedLayerNorm
    from easyparallellibrary_amd.ops.bias_gelu import FusedBiasGelu
    for _, _, child in hits:
        assert not isinstance(child, (FusedLayerNorm, FusedBiasGelu))


def test_auto_stage_repeated_block_policy():
    """Auto-stage must cut at repeated-block boundaries by STRUCTURE
    (reference planner.py:66-112 policy order), keeping the embedding
    prefix in stage 0 and the head in the last stage."""
    import torch.nn as nn
    from easyparallellibrary_amd.parallel.planner import AutoStageGenerator

    class Block(nn.Module):
        def __init__(self, h):
            super().__init__()
            self.fc = nn.Linear(h, h)

        def forward(self, x):
            return self.fc(x)

    h = 8
    spine = nn.Sequential(
        nn.Embedding(1000, h),            # heavy prefix
        *[Block(h) for _ in range(6)],
        nn.Linear(h, 1000),               # heavy head
    )
    stages = AutoStageGenerator(spine, 2).search()
    assert stages is not None and len(stages) == 2
    # every cut lands at a Block boundary: stage 0 starts with the
    # embedding, stage 1 starts with a Block (never mid-prefix/suffix)
    assert isinstance(stages[0][0], nn.Embedding)
    assert type(stages[1][0]).__name__ == "Block"
    assert isinstance(stages[1][-1], nn.Linear)
    assert sum(len(s) for s in stages) == len(spine)
    # the heavy embedding/head did NOT drag blocks unevenly: both stages
    # hold at least one Block
    n_blocks = [sum(1 for m in s if type(m).__name__ == "Block")
                for s in stages]
    assert all(n >= 1 for n in n_blocks), n_blocks
