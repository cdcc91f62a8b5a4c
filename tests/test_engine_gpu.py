"""Engine end-to-end on one MI355X: bf16 BERT tiny trains; fused kernels
are the executing path."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_bert_tiny_bf16_trains():
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.models import bert
    from easyparallellibrary_amd.ops.distributed_losses import (
        ParallelCrossEntropy)

    epl.init()
    torch.manual_seed(0)
    model = bert.build_bert(dict(layers=2, hidden=256, heads=4, ffn=1024),
                            vocab_size=2048, max_pos=128)
    loss_fn = ParallelCrossEntropy()
    engine = epl.Engine(model, loss_fn=loss_fn, optimizer="adamw", lr=3e-4,
                        dtype=torch.bfloat16)
    ids, tgt = bert.synthetic_mlm_batch(8, 128, 2048, device=engine.device,
                                        seed=3)
    losses = [float(engine.train_step(ids, tgt)) for _ in range(8)]
    torch.cuda.synchronize()
    assert all(torch.isfinite(torch.tensor(losses)))
    assert losses[-1] < losses[0], losses


def test_lamb_gpu_step_finite():
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.models import bert
    from easyparallellibrary_amd.ops.distributed_losses import (
        ParallelCrossEntropy)

    epl.init()
    model = bert.build_bert(dict(layers=1, hidden=128, heads=2, ffn=256),
                            vocab_size=512, max_pos=64)
    engine = epl.Engine(model, loss_fn=ParallelCrossEntropy(),
                        optimizer="lamb", lr=1e-3, dtype=torch.bfloat16)
    ids, tgt = bert.synthetic_mlm_batch(4, 64, 512, device=engine.device,
                                        seed=5)
    l0 = float(engine.train_step(ids, tgt))
    l1 = float(engine.train_step(ids, tgt))
    torch.cuda.synchronize()
    assert torch.isfinite(torch.tensor([l0, l1])).all()


def test_checkpoint_roundtrip_gpu(tmp_path):
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.models import bert
    from easyparallellibrary_amd.ops.distributed_losses import (
        ParallelCrossEntropy)
    epl.init()
    torch.manual_seed(3)
    model = bert.build_bert(dict(layers=1, hidden=128, heads=2, ffn=256),
                            vocab_size=512, max_pos=64)
    engine = epl.Engine(model, loss_fn=ParallelCrossEntropy(),
                        optimizer="adamw", lr=1e-3, dtype=torch.bfloat16)
    ids, tgt = bert.synthetic_mlm_batch(4, 64, 512, device=engine.device,
                                        seed=5)
    engine.train_step(ids, tgt)
    engine.save_checkpoint(str(tmp_path))
    saved_params = engine.flat_groups[0].param_arena.detach().clone()
    saved_master = engine.flat_groups[0].master_arena.detach().clone()
    cont = [float(engine.train_step(ids, tgt)) for _ in range(2)]

    from easyparallellibrary_amd.env import Env
    from easyparallellibrary_amd.parallel import hooks
    hooks.remove_hooks()
    Env._instance = None
    epl.init()
    torch.manual_seed(99)  # different init; restore must override
    model2 = bert.build_bert(dict(layers=1, hidden=128, heads=2, ffn=256),
                             vocab_size=512, max_pos=64)
    engine2 = epl.Engine(model2, loss_fn=ParallelCrossEntropy(),
                         optimizer="adamw", lr=1e-3, dtype=torch.bfloat16)
    engine2.load_checkpoint(str(tmp_path))
    # restore is exact: params and fp32 master match the snapshot
    assert torch.equal(engine2.flat_groups[0].param_arena, saved_params)
    assert torch.equal(engine2.flat_groups[0].master_arena, saved_master)
    resumed = [float(engine2.train_step(ids, tgt)) for _ in range(2)]
    torch.cuda.synchronize()
    # loss trajectory only loosely: bf16 + atomics-based attention
    # backward is not bit-deterministic across runs
    assert all(abs(a - b) < 5e-2 for a, b in zip(cont, resumed)), (
        cont, resumed)


def test_gpt2_tiny_gc_native_attention_trains():
    """Gradient checkpointing + native attention: recompute must not
    re-trigger saved-tensor unpacks (regression: ctx.saved_tensors
    accessed twice in the Function backward broke under
    torch.utils.checkpoint)."""
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.models import gpt2
    from easyparallellibrary_amd.ops.distributed_losses import (
        ParallelCrossEntropy)
    epl.init(epl.Config({"gradient_checkpoint.type": "auto"}))
    torch.manual_seed(0)
    model = gpt2.build_gpt2(dict(layers=3, hidden=256, heads=4, ffn=1024),
                            vocab_size=1024, max_pos=128)
    engine = epl.Engine(model, loss_fn=ParallelCrossEntropy(),
                        optimizer="adamw", lr=3e-4, dtype=torch.bfloat16)
    ids = torch.randint(0, 1024, (4, 128), device=engine.device)
    tgt = torch.randint(0, 1024, (4 * 128,), device=engine.device)
    losses = [float(engine.train_step(ids, tgt)) for _ in range(4)]
    torch.cuda.synchronize()
    assert all(torch.isfinite(torch.tensor(losses)))
    assert losses[-1] < losses[0], losses


def test_amp_o1_dynamic_scale_gpu():
    """AMP O1 (fp32 params + autocast + dynamic loss scale) trains on
    the GPU: loss decreases and the scaler stays finite."""
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.models import bert
    from easyparallellibrary_amd.ops.distributed_losses import (
        ParallelCrossEntropy)
    epl.init(epl.Config({"amp.level": "O1", "amp.dtype": "bf16"}))
    torch.manual_seed(2)
    model = bert.build_bert(dict(layers=2, hidden=256, heads=4, ffn=1024),
                            vocab_size=1024, max_pos=128)
    engine = epl.Engine(model, loss_fn=ParallelCrossEntropy(),
                        optimizer="adamw", lr=3e-4)  # fp32 + autocast
    ids, tgt = bert.synthetic_mlm_batch(8, 128, 1024,
                                        device=engine.device, seed=3)
    losses = [float(engine.train_step(ids, tgt)) for _ in range(6)]
    torch.cuda.synchronize()
    assert all(torch.isfinite(torch.tensor(losses)))
    assert losses[-1] < losses[0], losses


def test_grad_clip_gpu():
    """optimizer.max_grad_norm on GPU: the clipped step must stay
    finite and train."""
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.models import bert
    from easyparallellibrary_amd.ops.distributed_losses import (
        ParallelCrossEntropy)
    epl.init(epl.Config({"optimizer.max_grad_norm": 0.5}))
    torch.manual_seed(4)
    model = bert.build_bert(dict(layers=2, hidden=256, heads=4, ffn=1024),
                            vocab_size=1024, max_pos=128)
    engine = epl.Engine(model, loss_fn=ParallelCrossEntropy(),
                        optimizer="adamw", lr=3e-4, dtype=torch.bfloat16)
    ids, tgt = bert.synthetic_mlm_batch(8, 128, 1024,
                                        device=engine.device, seed=5)
    losses = [float(engine.train_step(ids, tgt)) for _ in range(6)]
    torch.cuda.synchronize()
    assert all(torch.isfinite(torch.tensor(losses)))
    assert losses[-1] < losses[0], losses


def test_ring_module_native_gpu():
    """RingSelfAttention at world 1 rides the native with_lse kernels
    (bf16, d64) and trains."""
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.ops.ring_attention import (
        RingSelfAttention)
    epl.init()
    torch.manual_seed(6)
    m = RingSelfAttention(256, 4, comm=None, causal=True).to(
        "cuda", torch.bfloat16)
    x = torch.randn(2, 512, 256, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    y = m(x)
    y.sum().backward()
    torch.cuda.synchronize()
    assert torch.isfinite(x.grad.float()).all()


def test_moe_native_dispatch_matches_torch_path():
    """The fused dispatch/combine kernels (csrc/kernels/moe.hip) must
    match the torch index path exactly-ish: same output, same x / gate
    / expert-weight grads (fp32-accumulated combine vs bf16 index_add
    gives small rounding differences)."""
    import os
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.ops import moe as moe_mod
    from easyparallellibrary_amd.ops.moe import ExpertParallelMLP
    epl.init()
    torch.manual_seed(3)

    def run(native):
        os.environ["EPL_MOE_NATIVE_DISPATCH"] = "1" if native else "0"
        torch.manual_seed(10)
        m = ExpertParallelMLP(512, 2048, 8).to("cuda", torch.bfloat16)
        x = torch.randn(4, 256, 512, device="cuda", dtype=torch.bfloat16,
                        requires_grad=True)
        y = m(x)
        y.float().pow(2).mean().backward()
        return (y.detach().float(), x.grad.float().clone(),
                m.gate.weight.grad.float().clone(),
                m.w1.grad.float().clone())

    try:
        yn, gxn, ggn, gw1n = run(True)
        yt, gxt, ggt, gw1t = run(False)
    finally:
        os.environ.pop("EPL_MOE_NATIVE_DISPATCH", None)
    torch.cuda.synchronize()
    assert torch.allclose(yn, yt, atol=3e-2), (yn - yt).abs().max()
    assert torch.allclose(gxn, gxt, atol=3e-2), (gxn - gxt).abs().max()
    assert torch.allclose(ggn, ggt, atol=3e-2,
                          rtol=3e-2), (ggn - ggt).abs().max()
    assert torch.allclose(gw1n, gw1t, atol=3e-2), (gw1n - gw1t).abs().max()


def test_gpt2_tiny_convergence_200_steps():
    """Convergence soak: 200 steps on a tiny GPT-2 (native causal
    attention + fused kernels) must drive the LM loss well below init
    — catches subtle gradient bugs single-step checks miss."""
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.models import gpt2
    from easyparallellibrary_amd.ops.distributed_losses import (
        ParallelCrossEntropy)
    epl.init()
    torch.manual_seed(1)
    model = gpt2.build_gpt2(dict(layers=2, hidden=256, heads=4, ffn=1024),
                            vocab_size=512, max_pos=128)
    engine = epl.Engine(model, loss_fn=ParallelCrossEntropy(),
                        optimizer="adamw", lr=3e-4, dtype=torch.bfloat16)
    # one small fixed batch: the model must be able to (over)fit it
    ids, tgt = gpt2.synthetic_lm_batch(4, 128, 512, device=engine.device,
                                       seed=5)
    losses = [float(engine.train_step(ids, tgt)) for _ in range(200)]
    torch.cuda.synchronize()
    assert all(torch.isfinite(torch.tensor(losses)))
    import math
    assert losses[-1] < math.log(512) * 0.35, (losses[0], losses[-1])


def test_engine_collectives_api_gpu():
    """write_summaries / merged_collections / all_reduce_metric /
    broadcast_signal at world 1 on GPU."""
    import easyparallellibrary_amd as epl
    epl.init()
    import torch.nn as nn
    with epl.replicate(device_count=1):
        model = nn.Linear(8, 2)
    engine = epl.Engine(model, loss_fn=nn.MSELoss(),
                        dtype=torch.bfloat16)
    epl.add_to_collection(1.5, epl.GraphKeys.GLOBAL_MEAN_OBJECTS)
    merged = engine.merged_collections()
    assert float(merged[epl.GraphKeys.GLOBAL_MEAN_OBJECTS][0]) == 1.5
    v = engine.all_reduce_metric(torch.tensor(2.0, device=engine.device))
    assert float(v) == 2.0
    assert engine.broadcast_signal(3.0) == 3.0

    class W:
        seen = {}

        def add_scalar(self, n, val, s):
            self.seen[n] = val
    w = W()
    out = engine.write_summaries(w, 0, scalars={"x": 4.0})
    assert out["x"] == 4.0 and w.seen["x"] == 4.0
    engine.close()
