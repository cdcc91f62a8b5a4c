"""hipGraph-captured training step on GPU (runtime/hipgraph.py):
capture really happens, replayed steps train the same model the eager
path does, and shape changes fail loudly."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def _run(graphed, steps=8):
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.models.moe_transformer import (
        build_moe_transformer)
    from easyparallellibrary_amd.ops.distributed_losses import (
        ParallelCrossEntropy)

    epl.init(epl.Config({"kernel.hip_graph": graphed,
                         "cluster.colocate_split_and_replicate": True}))
    torch.manual_seed(5)
    model = build_moe_transformer(world=1, layers=2, hidden=128, heads=2,
                                  ffn=256, num_experts=4, vocab_size=512,
                                  max_pos=64)
    engine = epl.Engine(model, loss_fn=ParallelCrossEntropy(),
                        optimizer="adamw", lr=1e-3, dtype=torch.bfloat16)
    torch.manual_seed(11)
    ids = torch.randint(0, 512, (4, 64), device=engine.device)
    tgt = torch.randint(0, 512, (4, 64), device=engine.device)
    losses = [float(engine.train_step(ids, tgt)) for _ in range(steps)]
    torch.cuda.synchronize()
    return engine, losses


def test_hipgraph_step_matches_eager():
    engine_g, losses_g = _run(True)
    # capture really happened (3 eager warmups, then capture + replays)
    from easyparallellibrary_amd.runtime.hipgraph import HipGraphStep
    assert isinstance(engine_g._hipgraph, HipGraphStep)
    assert engine_g._hipgraph.graph is not None

    engine_e, losses_e = _run(False)
    assert engine_e._hipgraph is False
    # same model/data/seeds: the replayed graph must train the model the
    # eager path does (bf16 + fused-GEMM accumulation order leaves a
    # small numeric slack)
    for lg, le in zip(losses_g, losses_e):
        assert abs(lg - le) <= 0.05 * abs(le) + 1e-2, (losses_g, losses_e)
    assert losses_g[-1] < losses_g[0]


def test_hipgraph_shape_change_raises():
    engine, _ = _run(True, steps=5)
    ids = torch.randint(0, 512, (2, 64), device=engine.device)
    tgt = torch.randint(0, 512, (2, 64), device=engine.device)
    with pytest.raises(RuntimeError, match="hipGraph step captured"):
        engine.train_step(ids, tgt)


def test_hipgraph_with_gradient_checkpoint_matches_eager():
    """GC recompute runs inside the captured backward (dropout-free
    wrappers skip the RNG save/restore): graphed GC must train like
    eager GC."""
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.models import gpt2
    from easyparallellibrary_amd.ops.distributed_losses import (
        ParallelCrossEntropy)

    def run(graphed):
        epl.init(epl.Config({"kernel.hip_graph": graphed,
                             "gradient_checkpoint.type": "auto"}))
        torch.manual_seed(5)
        model = gpt2.build_gpt2(dict(layers=3, hidden=128, heads=2,
                                     ffn=512), vocab_size=512, max_pos=64)
        engine = epl.Engine(model, loss_fn=ParallelCrossEntropy(),
                            optimizer="adamw", lr=1e-3,
                            dtype=torch.bfloat16)
        assert engine._gc_wrapped
        torch.manual_seed(11)
        ids = torch.randint(0, 512, (4, 64), device=engine.device)
        tgt = torch.randint(0, 512, (4 * 64,), device=engine.device)
        losses = [float(engine.train_step(ids, tgt)) for _ in range(8)]
        torch.cuda.synchronize()
        return engine, losses

    engine_g, losses_g = run(True)
    from easyparallellibrary_amd.runtime.hipgraph import HipGraphStep
    assert isinstance(engine_g._hipgraph, HipGraphStep)
    assert engine_g._hipgraph.graph is not None
    _, losses_e = run(False)
    for lg, le in zip(losses_g, losses_e):
        assert abs(lg - le) <= 0.05 * abs(le) + 1e-2, (losses_g, losses_e)
    assert losses_g[-1] < losses_g[0]
