"""ResNet DP+TP-classifier config (BASELINE config 4 shape) and CPU-offload
optimizer, on CPU."""

import torch
import torch.nn as nn

from tests.utils import run_multiprocess


def _resnet_tp_worker(rank, world):
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.models.resnet import (
        build_resnet50_split_classifier, synthetic_image_batch)
    from easyparallellibrary_amd.ops import bridging
    from easyparallellibrary_amd.ops.distributed_losses import (
        ParallelCrossEntropy)

    epl.init(epl.Config({"cluster.colocate_split_and_replicate": True}))
    torch.manual_seed(70)
    # tiny resnet for CPU: 1 block per stage group, 200 classes
    from easyparallellibrary_amd.models import resnet as R
    with epl.replicate(world, name="backbone"):
        backbone = R.ResNetBackbone(layers=(1, 1, 1, 1), width=8)
    with epl.split(world, name="classifier"):
        head = nn.Linear(backbone.out_features, 200)
    model = R.ResNetClassifier(backbone, head)
    state = {}

    def loss_fn(logits, targets):
        comm = state["engine"].tp_comm
        full_t = bridging.replica_to_split(targets, comm)
        ce = ParallelCrossEntropy(comm=comm,
                                  vocab_begin=state["head"].offset)
        return ce(logits, full_t)

    engine = epl.Engine(model, loss_fn=loss_fn, optimizer="adamw", lr=1e-3)
    state["engine"] = engine
    state["head"] = model.head
    x, y = synthetic_image_batch(2, 200, size=32, seed=100 + rank)
    losses = [float(engine.train_step(x, y)) for _ in range(2)]
    return losses


def test_resnet_dp_tp_classifier():
    res = run_multiprocess(_resnet_tp_worker, world=2, timeout=300)
    assert res[0] == res[1]
    assert all(torch.isfinite(torch.tensor(res[0])))


def _offload_worker(rank, world):
    import easyparallellibrary_amd as epl
    epl.init(epl.Config({"offload.level": "v0"}))
    torch.manual_seed(80)
    with epl.replicate(1):
        model = nn.Sequential(nn.Linear(8, 16), nn.Tanh(), nn.Linear(16, 2))
    engine = epl.Engine(model, loss_fn=nn.MSELoss(), optimizer="adamw",
                        lr=1e-2)
    torch.manual_seed(81)
    x = torch.randn(4, 8)
    y = torch.randn(4, 2)
    return [float(engine.train_step(x, y)) for _ in range(4)]


def _no_offload_worker(rank, world):
    import easyparallellibrary_amd as epl
    epl.init()
    torch.manual_seed(80)
    with epl.replicate(1):
        model = nn.Sequential(nn.Linear(8, 16), nn.Tanh(), nn.Linear(16, 2))
    engine = epl.Engine(model, loss_fn=nn.MSELoss(), optimizer="adamw",
                        lr=1e-2)
    torch.manual_seed(81)
    x = torch.randn(4, 8)
    y = torch.randn(4, 2)
    return [float(engine.train_step(x, y)) for _ in range(4)]


def test_offload_matches_plain():
    off = run_multiprocess(_offload_worker, world=1)[0]
    plain = run_multiprocess(_no_offload_worker, world=1)[0]
    assert all(abs(a - b) < 1e-6 for a, b in zip(off, plain))
