"""Tensor-parallel (split) path on CPU/gloo: sharded linear + sharded-vocab
softmax CE must match the serial full model (reference: tests/split_test.py
semantics)."""

import torch
import torch.nn as nn

from tests.utils import run_multiprocess

B, IN, HID, NCLS = 4, 8, 16, 10


class Backbone(nn.Module):
    def __init__(self):
        super().__init__()
        self.fc = nn.Linear(IN, HID)

    def forward(self, x):
        return torch.tanh(self.fc(x))


def _tp_worker(rank, world):
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.ops import bridging
    from easyparallellibrary_amd.ops.distributed_losses import (
        ParallelCrossEntropy)

    epl.init(epl.Config({"cluster.colocate_split_and_replicate": True}))
    torch.manual_seed(42)
    with epl.replicate(world, name="backbone"):
        backbone = Backbone()
    with epl.split(world, name="head"):
        head = nn.Linear(HID, NCLS)

    class M(nn.Module):
        def __init__(self):
            super().__init__()
            self.backbone, self.head = backbone, head

        def forward(self, x):
            return self.head(self.backbone(x))

    model = M()
    state = {}

    def loss_fn(logits, targets):
        comm = state["engine"].tp_comm
        full_targets = bridging.replica_to_split(targets, comm)
        ce = ParallelCrossEntropy(
            comm=comm, vocab_begin=state["head"].offset)
        return ce(logits, full_targets)

    engine = epl.Engine(model, loss_fn=loss_fn, optimizer="adamw", lr=1e-2)
    state["engine"] = engine
    state["head"] = model.head  # now a ColumnParallelLinear
    torch.manual_seed(100 + rank)
    x = torch.randn(B, IN)
    y = torch.randint(0, NCLS, (B,))
    losses = [float(engine.train_step(x, y)) for _ in range(3)]
    return losses


def _serial_worker(rank, world):
    import easyparallellibrary_amd as epl
    epl.init()
    torch.manual_seed(42)
    with epl.replicate(1):
        backbone = Backbone()
        head = nn.Linear(HID, NCLS)

    class M(nn.Module):
        def __init__(self):
            super().__init__()
            self.backbone, self.head = backbone, head

        def forward(self, x):
            return self.head(self.backbone(x))

    engine = epl.Engine(M(), loss_fn=nn.CrossEntropyLoss(), optimizer="adamw",
                        lr=1e-2)
    torch.manual_seed(100)
    x0 = torch.randn(B, IN)
    y0 = torch.randint(0, NCLS, (B,))
    torch.manual_seed(101)
    x1 = torch.randn(B, IN)
    y1 = torch.randint(0, NCLS, (B,))
    x = torch.cat([x0, x1])
    y = torch.cat([y0, y1])
    return [float(engine.train_step(x, y)) for _ in range(3)]


def test_tp2_matches_serial_full_batch():
    serial = run_multiprocess(_serial_worker, world=1)[0]
    tp = run_multiprocess(_tp_worker, world=2)
    assert tp[0] == tp[1]  # same full-batch loss on both shards
    for a, b in zip(serial, tp[0]):
        assert abs(a - b) < 1e-4, (serial, tp[0])


def test_shard_sizes_remainder():
    from easyparallellibrary_amd.ops.distributed_dense import (
        shard_offset, shard_size)
    # remainder goes to shard 0 (reference distributed_dense.py:102-108)
    assert [shard_size(10, 3, s) for s in range(3)] == [4, 3, 3]
    assert [shard_offset(10, 3, s) for s in range(3)] == [0, 4, 7]


def _tp_embedding_worker(rank, world):
    """Vocab-sharded embedding inside a split scope matches the full
    embedding."""
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.ops.distributed_losses import (
        ParallelCrossEntropy)
    epl.init(epl.Config({"cluster.colocate_split_and_replicate": True}))
    torch.manual_seed(60)
    with epl.replicate(world, name="trunk"):
        trunk = nn.Linear(8, 8)
    with epl.split(world, name="emb"):
        emb = nn.Embedding(11, 8)   # odd vocab: remainder shard policy

    class M(nn.Module):
        def __init__(self):
            super().__init__()
            self.emb, self.trunk = emb, trunk

        def forward(self, ids):
            return self.trunk(torch.tanh(self.emb(ids)))

    engine = epl.Engine(M(), loss_fn=nn.MSELoss(), optimizer="adamw",
                        lr=1e-2)
    torch.manual_seed(61)
    ids = torch.randint(0, 11, (4, 6))
    y = torch.randn(4, 6, 8)
    return [float(engine.train_step(ids, y)) for _ in range(3)]


def test_tp_embedding_sharded():
    res = run_multiprocess(_tp_embedding_worker, world=2)
    assert res[0] == res[1]
    assert res[0][-1] < res[0][0]


def _argmax_worker(rank, world):
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.comm.backend import create_communicator
    from easyparallellibrary_amd.env import Env
    from easyparallellibrary_amd.ops.distributed_ops import (
        distributed_argmax, distributed_equal)
    epl.init()
    Env.get().get_or_create_process_group()
    comm = create_communicator("am", list(range(world)))
    torch.manual_seed(70)
    full = torch.randn(6, 10)          # same on every rank
    shard = 10 // world
    lo = rank * shard
    local = full[:, lo:lo + shard]
    pred = distributed_argmax(local, comm, vocab_begin=lo)
    eq = distributed_equal(pred, full.argmax(dim=-1))
    return pred, bool(eq.all())


def test_distributed_argmax_matches_full():
    res = run_multiprocess(_argmax_worker, world=2)
    assert res[0][1] and res[1][1]
    assert torch.equal(res[0][0], res[1][0])


def test_distributed_glorot_full_fan_variance():
    from easyparallellibrary_amd.ops.initializers import (
        distributed_glorot_uniform_)
    import math
    g = torch.Generator().manual_seed(0)
    shard = torch.empty(64, 256)           # half of a [128, 256] layer
    distributed_glorot_uniform_(shard, full_fan_in=256, full_fan_out=128,
                                generator=g)
    limit = math.sqrt(6.0 / (256 + 128))
    assert shard.abs().max() <= limit
    # variance of U(-l, l) is l^2/3; sampled variance within 10%
    assert abs(shard.var().item() - limit ** 2 / 3) < 0.1 * limit ** 2 / 3


def test_shard_roundtrip_randomized():
    """shard_offset/shard_size tile [0, n) exactly for random (n, w)."""
    import random
    from easyparallellibrary_amd.ops.distributed_dense import (
        shard_offset, shard_size)
    rng = random.Random(3)
    for _ in range(200):
        w = rng.randint(1, 9)
        n = rng.randint(w, 10000)
        pos = 0
        for s in range(w):
            assert shard_offset(n, w, s) == pos
            sz = shard_size(n, w, s)
            assert sz >= n // w
            pos += sz
        assert pos == n


def _tp_mlp_worker(rank, world):
    """Megatron-style column->row MLP pair: full output on every rank
    after ONE all-reduce; matches the serial MLP exactly."""
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.comm.backend import create_communicator
    from easyparallellibrary_amd.env import Env
    from easyparallellibrary_amd.ops.tp_mlp import TensorParallelMLP
    epl.init()
    Env.get().get_or_create_process_group()
    comm = create_communicator("tpmlp", list(range(world)))
    torch.manual_seed(21)
    fc1 = nn.Linear(16, 32)
    fc2 = nn.Linear(32, 16)
    tp = TensorParallelMLP(16, 32, comm, act=nn.GELU(),
                           source_fc1=fc1, source_fc2=fc2)
    torch.manual_seed(22)
    x = torch.randn(4, 16, requires_grad=True)
    out = tp(x)
    out.square().sum().backward()

    xr = x.detach().clone().requires_grad_(True)
    ref = fc2(nn.functional.gelu(fc1(xr)))
    ref.square().sum().backward()
    return ((out - ref).abs().max().item(),
            (x.grad - xr.grad).abs().max().item())


def test_tensor_parallel_mlp_pair():
    res = run_multiprocess(_tp_mlp_worker, world=2)
    for fwd_err, dx_err in res:
        assert fwd_err < 1e-5, fwd_err
        assert dx_err < 1e-4, dx_err


class _PlainQKVAttention(nn.Module):
    """Serial reference with separate q/k/v projections (the layout
    TensorParallelSelfAttention shards)."""

    def __init__(self, hidden, heads, causal):
        super().__init__()
        self.q = nn.Linear(hidden, hidden)
        self.k = nn.Linear(hidden, hidden)
        self.v = nn.Linear(hidden, hidden)
        self.proj = nn.Linear(hidden, hidden)
        self.heads = heads
        self.d = hidden // heads
        self.causal = causal

    def forward(self, x):
        b, s, _ = x.shape
        shp = (b, s, self.heads, self.d)
        q = self.q(x).reshape(shp).transpose(1, 2)
        k = self.k(x).reshape(shp).transpose(1, 2)
        v = self.v(x).reshape(shp).transpose(1, 2)
        o = torch.nn.functional.scaled_dot_product_attention(
            q, k, v, is_causal=self.causal)
        return self.proj(o.transpose(1, 2).reshape(b, s, -1))


def _tp_attn_worker(rank, world, causal):
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.comm.backend import create_communicator
    from easyparallellibrary_amd.env import Env
    from easyparallellibrary_amd.ops.tp_mlp import (
        TensorParallelSelfAttention)
    epl.init()
    Env.get().get_or_create_process_group()
    comm = create_communicator("tpattn", list(range(world)))
    torch.manual_seed(41)
    ref = _PlainQKVAttention(32, 4, causal)
    tp = TensorParallelSelfAttention(32, 4, comm, causal=causal,
                                     source=ref)
    torch.manual_seed(42)
    x = torch.randn(2, 8, 32, requires_grad=True)
    out = tp(x)
    out.square().sum().backward()
    xr = x.detach().clone().requires_grad_(True)
    r = ref(xr)
    r.square().sum().backward()
    return ((out - r).abs().max().item(),
            (x.grad - xr.grad).abs().max().item())


def test_tensor_parallel_attention():
    for causal in (False, True):
        res = run_multiprocess(_tp_attn_worker, world=2, args=(causal,))
        for fwd_err, dx_err in res:
            assert fwd_err < 1e-5, (causal, fwd_err)
            assert dx_err < 1e-4, (causal, dx_err)


def _dense_tp_engine_worker(rank, world):
    """Dense Megatron-TP block inside an epl.split scope, engine-driven:
    deferred sharding via the split transform's set_comm path."""
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.ops.tp_mlp import (
        TensorParallelMLP, TensorParallelSelfAttention)
    epl.init(epl.Config({"cluster.colocate_split_and_replicate": True}))
    torch.manual_seed(51)

    class Block(nn.Module):
        def __init__(self):
            super().__init__()
            self.ln1 = nn.LayerNorm(32)
            self.attn = TensorParallelSelfAttention(32, 4)
            self.ln2 = nn.LayerNorm(32)
            self.mlp = TensorParallelMLP(32, 64)

        def forward(self, x):
            x = x + self.attn(self.ln1(x))
            return x + self.mlp(self.ln2(x))

    with epl.replicate(world, name="trunk"):
        emb = nn.Linear(8, 32)
    with epl.split(world, name="tp"):
        block = Block()

    class M(nn.Module):
        def __init__(self):
            super().__init__()
            self.emb, self.block = emb, block
            self.head = nn.Linear(32, 4)

        def forward(self, x):
            return self.head(self.block(self.emb(x)))

    engine = epl.Engine(M(), loss_fn=nn.MSELoss(), optimizer="adamw",
                        lr=1e-2)
    torch.manual_seed(52)
    x = torch.randn(2, 8, 8)
    y = torch.randn(2, 8, 4)
    return [float(engine.train_step(x, y)) for _ in range(3)]


def test_dense_tp_block_in_engine_matches_serial():
    serial = _dense_tp_engine_worker(0, 1)
    from easyparallellibrary_amd.env import Env
    from easyparallellibrary_amd.parallel import hooks
    hooks.remove_hooks()
    Env._instance = None
    tp2 = run_multiprocess(_dense_tp_engine_worker, world=2)
    assert tp2[0] == tp2[1]
    for a, b in zip(serial, tp2[0]):
        assert abs(a - b) < 1e-5, (serial, tp2[0])


def _pp_tp_worker(rank, world, tp):
    """2 pipeline stages x width-tp stages of dense Megatron-TP blocks:
    stage outputs are full tensors, so position-wise p2p is exact."""
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.ops.tp_mlp import (
        TensorParallelMLP, TensorParallelSelfAttention)
    epl.init(epl.Config({
        "cluster.colocate_split_and_replicate": True,
        "pipeline.num_micro_batch": 2,
    }))
    torch.manual_seed(61)

    class Block(nn.Module):
        def __init__(self):
            super().__init__()
            self.ln1 = nn.LayerNorm(32)
            self.attn = TensorParallelSelfAttention(32, 4, causal=True)
            self.ln2 = nn.LayerNorm(32)
            self.mlp = TensorParallelMLP(32, 64)

        def forward(self, x):
            x = x + self.attn(self.ln1(x))
            return x + self.mlp(self.ln2(x))

    class Stage(nn.Module):
        def __init__(self, first, last):
            super().__init__()
            self.inp = nn.Linear(8, 32) if first else None
            self.out = nn.Linear(32, 4) if last else None
            self.blocks = nn.ModuleList()

        def forward(self, x):
            if self.inp is not None:
                x = self.inp(x)
            for b in self.blocks:
                x = b(x)
            if self.out is not None:
                x = self.out(x)
            return x

    stages = []
    for s in range(2):
        with epl.replicate(tp, name="stage_{}".format(s)):
            st = Stage(first=(s == 0), last=(s == 1))
            stages.append(st)
        with epl.split(tp, name="tp_{}".format(s)):
            st.blocks.append(Block())

    class M(nn.Module):
        def __init__(self):
            super().__init__()
            self.stages = nn.ModuleList(stages)

        def forward(self, x):
            for st in self.stages:
                x = st(x)
            return x

    engine = epl.Engine(M(), loss_fn=nn.MSELoss(), optimizer="adamw",
                        lr=1e-2)
    torch.manual_seed(62)
    x = torch.randn(4, 8, 8)
    y = torch.randn(4, 8, 4)
    out = []
    for _ in range(3):
        loss = engine.train_step(x, y)
        out.append(None if loss is None else float(loss))
    return out


def test_pp2_dense_tp2_hybrid():
    """PP2 x dense-TP2 on 4 ranks matches plain PP2 on 2 ranks."""
    base = run_multiprocess(_pp_tp_worker, world=2, args=(1,))
    hyb = run_multiprocess(_pp_tp_worker, world=4, args=(2,),
                           timeout=300)
    # last-stage positions: ranks 2 and 3
    assert hyb[2] == hyb[3]
    for a, b in zip(base[1], hyb[2]):
        assert abs(a - b) < 1e-5, (base[1], hyb[2])


def _tp_pipeline_builder_worker(rank, world, tp, strategy=None):
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.models.tp_transformer import (
        build_tp_pipeline)
    cfg = {
        "cluster.colocate_split_and_replicate": True,
        # 4 micro-batches so 1F1B's steady-state loop (fused per-link
        # exchanges) runs repeatedly, not just warmup/cooldown
        "pipeline.num_micro_batch": 4,
    }
    if strategy:
        cfg["pipeline.strategy"] = strategy
    epl.init(epl.Config(cfg))
    torch.manual_seed(71)
    model = build_tp_pipeline(stages=2, tp=tp, layers=2, hidden=32,
                              heads=4, ffn=64, vocab_size=128,
                              max_pos=32)

    def lm_loss(logits, targets):
        return nn.functional.cross_entropy(logits.reshape(-1, 128),
                                           targets)

    engine = epl.Engine(model, loss_fn=lm_loss, optimizer="adamw",
                        lr=1e-3)
    torch.manual_seed(72)
    ids = torch.randint(0, 128, (4, 16))
    tgt = torch.randint(0, 128, (4 * 16,))
    out = []
    for _ in range(3):
        loss = engine.train_step(ids, tgt)
        out.append(None if loss is None else float(loss))
    return out


def test_build_tp_pipeline_matches_plain_pp():
    base = run_multiprocess(_tp_pipeline_builder_worker, world=2,
                            args=(1,))
    hyb = run_multiprocess(_tp_pipeline_builder_worker, world=4,
                           args=(2,), timeout=300)
    assert hyb[2] == hyb[3]
    for a, b in zip(base[1], hyb[2]):
        assert abs(a - b) < 1e-5, (base[1], hyb[2])


def test_mixed_width_tp_pipeline_matches_plain_pp():
    """Stage widths [1, 2]: a plain width-1 stage feeding a dense-TP-2
    stage (pipeline runtime fans the activation out to both positions
    and takes the already-all-reduced grad from position 0).  Losses
    must match the all-width-1 pipeline exactly."""
    base = run_multiprocess(_tp_pipeline_builder_worker, world=2,
                            args=(1,))
    mixed = run_multiprocess(_tp_pipeline_builder_worker, world=3,
                             args=([1, 2],), timeout=300)
    # last stage spans ranks 1 and 2 — identical replicated losses
    assert mixed[1] == mixed[2]
    for a, b in zip(base[1], mixed[1]):
        assert abs(a - b) < 1e-5, (base[1], mixed[1])


def test_mixed_width_tp_pipeline_dp2():
    """DP x mixed-width hybrid: 2 replicas of a [1, 2]-width pipeline on
    6 ranks (exercises the multi-replica boundary-link creation in
    _init_mixed).  Same seed => both replicas see identical data, so the
    DP allreduce is an identity and losses must still match plain PP2."""
    base = run_multiprocess(_tp_pipeline_builder_worker, world=2,
                            args=(1,))
    dp2 = run_multiprocess(_tp_pipeline_builder_worker, world=6,
                           args=([1, 2],), timeout=300)
    # last stage of replica 0 = ranks 1,2; of replica 1 = ranks 4,5
    assert dp2[1] == dp2[2] == dp2[4] == dp2[5]
    for a, b in zip(base[1], dp2[1]):
        assert abs(a - b) < 1e-5, (base[1], dp2[1])


def test_mixed_width_tp_pipeline_wide_first():
    """Stage widths [2, 1]: the k->1 direction — position 0 sends the
    replicated stage output; the narrow rank fans the grad back to both
    positions (the 'reduce' op's backward is identity per rank)."""
    base = run_multiprocess(_tp_pipeline_builder_worker, world=2,
                            args=(1,))
    mixed = run_multiprocess(_tp_pipeline_builder_worker, world=3,
                             args=([2, 1],), timeout=300)
    for a, b in zip(base[1], mixed[2]):
        assert abs(a - b) < 1e-5, (base[1], mixed[2])


def _auto_pair_worker(rank, world):
    """auto.auto_pair_sequential: plain Linears inside an nn.Sequential
    under a split scope become a Megatron column->row pair (one
    all-reduce), numerically identical to serial training."""
    import easyparallellibrary_amd as epl
    epl.init(epl.Config({"cluster.colocate_split_and_replicate": True,
                         "auto.auto_pair_sequential": True}))
    torch.manual_seed(61)
    with epl.replicate(world, name="trunk"):
        emb = nn.Linear(8, 16)
    with epl.split(world, name="tp"):
        mlp = nn.Sequential(nn.Linear(16, 32), nn.GELU(),
                            nn.Linear(32, 16))

    class M(nn.Module):
        def __init__(self):
            super().__init__()
            self.emb, self.mlp = emb, mlp
            self.head = nn.Linear(16, 4)

        def forward(self, x):
            return self.head(self.mlp(self.emb(x)))

    engine = epl.Engine(M(), loss_fn=nn.MSELoss(), optimizer="adamw",
                        lr=1e-2)
    if world > 1:
        from easyparallellibrary_amd.ops.split_transform import (
            PairedColumnLinear)
        from easyparallellibrary_amd.ops.distributed_dense import (
            RowParallelLinear)
        assert isinstance(mlp[0], PairedColumnLinear), type(mlp[0])
        assert isinstance(mlp[2], RowParallelLinear), type(mlp[2])
    torch.manual_seed(62)
    x = torch.randn(2, 8, 8)
    y = torch.randn(2, 8, 4)
    return [float(engine.train_step(x, y)) for _ in range(3)]


def test_auto_pair_sequential_matches_serial():
    serial = _auto_pair_worker(0, 1)
    from easyparallellibrary_amd.env import Env
    from easyparallellibrary_amd.parallel import hooks
    hooks.remove_hooks()
    Env._instance = None
    tp2 = run_multiprocess(_auto_pair_worker, world=2)
    assert tp2[0] == tp2[1]
    for a, b in zip(serial, tp2[0]):
        assert abs(a - b) < 1e-5, (serial, tp2[0])


def _tp3_worker(rank, world, tp):
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.models.tp_transformer import (
        build_tp_pipeline)
    epl.init(epl.Config({
        "cluster.colocate_split_and_replicate": True,
        "pipeline.num_micro_batch": 4,
    }))
    torch.manual_seed(73)
    model = build_tp_pipeline(stages=3, tp=tp, layers=3, hidden=32,
                              heads=4, ffn=64, vocab_size=128,
                              max_pos=32)

    def lm_loss(logits, targets):
        return nn.functional.cross_entropy(logits.reshape(-1, 128),
                                           targets)

    engine = epl.Engine(model, loss_fn=lm_loss, optimizer="adamw",
                        lr=1e-3)
    torch.manual_seed(74)
    ids = torch.randint(0, 128, (4, 16))
    tgt = torch.randint(0, 128, (4 * 16,))
    return [float(l) if (l := engine.train_step(ids, tgt)) is not None
            else None for _ in range(3)]


def test_mixed_width_equal_wide_boundary():
    """Widths [1, 2, 2]: covers the wide<->wide equal boundary inside a
    mixed pipeline (replicated tensors on both per-position links) on
    top of the 1->k entry boundary."""
    base = run_multiprocess(_tp3_worker, world=3, args=([1, 1, 1],),
                            timeout=300)
    mixed = run_multiprocess(_tp3_worker, world=5, args=([1, 2, 2],),
                             timeout=300)
    assert mixed[3] == mixed[4]
    for a, b in zip(base[2], mixed[3]):
        assert abs(a - b) < 1e-5, (base[2], mixed[3])


def test_mixed_width_gpipe_schedule():
    """Mixed widths under the PreferForward (GPipe) schedule (the
    default-schedule tests above exercise the per-link 1F1B path)."""
    base = run_multiprocess(_tp_pipeline_builder_worker, world=2,
                            args=(1,))
    mixed = run_multiprocess(_tp_pipeline_builder_worker, world=3,
                             args=([1, 2], "prefer_forward"),
                             timeout=300)
    assert mixed[1] == mixed[2]
    for a, b in zip(base[1], mixed[1]):
        assert abs(a - b) < 1e-5, (base[1], mixed[1])


def _unmarked_mixed_worker(rank, world):
    """Mixed widths where the wide stage is NOT replicated_io: the
    pipeline runtime must refuse (EP-style stages carry per-position
    data streams and cannot change width)."""
    import easyparallellibrary_amd as epl
    epl.init(epl.Config({"pipeline.num_micro_batch": 2}))
    torch.manual_seed(81)
    with epl.replicate(1, name="stage_0"):
        s0 = nn.Linear(8, 16)
    with epl.replicate(2, name="stage_1"):  # wide but unmarked
        s1 = nn.Linear(16, 4)

    class M(nn.Module):
        def __init__(self):
            super().__init__()
            self.s0, self.s1 = s0, s1

        def forward(self, x):
            return self.s1(self.s0(x))

    try:
        epl.Engine(M(), loss_fn=nn.MSELoss())
    except ValueError as e:
        return "replicated_io" in str(e)
    return False


def test_mixed_width_requires_replicated_io():
    """An unmarked wide stage next to a narrow one must be refused —
    both at the TaskGraph level and when the engine builds the runtime."""
    from easyparallellibrary_amd.ir.plan import TaskGraph
    from easyparallellibrary_amd.strategies.replicate import Replicate
    tg_w = TaskGraph(1, Replicate(2, name="s1"))
    assert not tg_w.replicated_io
    tg_ok = TaskGraph(2, Replicate(2, name="s2", replicated_io=True))
    assert tg_ok.replicated_io
    out = run_multiprocess(_unmarked_mixed_worker, world=3, timeout=300)
    assert all(out), out


def _mixed_eval_worker(rank, world, widths):
    """Pipelined EVAL across mixed-width stages: the 1<->k eval fan-out
    and fan-in must follow the same link rules as training."""
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.models.tp_transformer import (
        build_tp_pipeline)
    epl.init(epl.Config({
        "cluster.colocate_split_and_replicate": True,
        "pipeline.num_micro_batch": 2,
    }))
    torch.manual_seed(71)
    model = build_tp_pipeline(stages=2, tp=(widths or 1), layers=2,
                              hidden=32, heads=4, ffn=64, vocab_size=128,
                              max_pos=32)
    engine = epl.Engine(model, loss_fn=lambda o, t: o.sum())
    torch.manual_seed(72)
    ids = torch.randint(0, 128, (4, 16))
    out = engine.eval_step(ids)
    engine.close()
    return None if out is None else out.float().clone()


def test_mixed_width_pipelined_eval_matches_serial():
    serial = run_multiprocess(_mixed_eval_worker, world=2,
                              args=(None,))  # plain [1, 1]
    mixed = run_multiprocess(_mixed_eval_worker, world=3,
                             args=([1, 2],), timeout=300)
    # last stage of the mixed run is wide: BOTH positions return the
    # (replicated) outputs, identical to the plain-PP2 run
    assert serial[0] is None and serial[1] is not None
    assert mixed[0] is None
    for pos_out in (mixed[1], mixed[2]):
        assert pos_out is not None
        assert torch.allclose(pos_out, serial[1], atol=1e-5), (
            (pos_out - serial[1]).abs().max())
