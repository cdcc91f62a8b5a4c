#!/usr/bin/env python3
"""Benchmark harness (driver contract — see BASELINE.json).

Headline metric: samples/sec (whole node) for BERT-Large DP(+PP) on
synthetic MLM data, bf16, random-init weights (BASELINE.json configs).

  python bench.py --gpus N --steps K --warmup W
      [--config bert_dp|bert_pp|bert_zero|resnet_tp|gpt2_xl|moe|moe_pp]
      [--pp S] [--micro-batch M] [--batch B] [--zero v0|v1]

For N>1 the driver launches this under torch.distributed.run with one rank
per GPU (RCCL over xGMI); ranks read RANK/LOCAL_RANK/WORLD_SIZE from env.
Weak scaling: per-GPU batch fixed as N grows.
"""

import argparse
import json
import os
import time

import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--config", default="bert_dp",
                   choices=["bert_dp", "bert_pp", "bert_zero", "resnet_tp",
                            "gpt2_xl", "moe", "moe_pp", "tp_pp"])
    p.add_argument("--model", default=None)
    p.add_argument("--batch", type=int, default=0,
                   help="per-GPU samples per step (0 = config default)")
    p.add_argument("--seq-len", type=int, default=512)
    p.add_argument("--pp", type=int, default=0, help="pipeline stages")
    p.add_argument("--micro-batch", type=int, default=0)
    p.add_argument("--zero", default=None, choices=[None, "", "v0", "v1"])
    p.add_argument("--gc", default=None, choices=[None, "", "auto"])
    p.add_argument("--offload", default=None, choices=[None, "", "v0"])
    p.add_argument("--layers", type=int, default=0,
                   help="override model depth (smoke runs)")
    p.add_argument("--hidden", type=int, default=0,
                   help="override hidden size (smoke runs)")
    p.add_argument("--device", default=None)
    p.add_argument("--dropout", type=float, default=0.0,
                   help="attention dropout (runs in the native kernels)")
    return p.parse_args()


def build_bert_bench(args, epl, world, on_gpu, dtype):
    from easyparallellibrary_amd.models import bert
    from easyparallellibrary_amd.ops.distributed_losses import (
        ParallelCrossEntropy)
    batch = args.batch or 128
    pp = args.pp or (2 if args.config == "bert_pp" else 1)
    if world % max(pp, 1) != 0 or world < pp:
        pp = 1  # degrade gracefully (e.g. bert_pp on 1 GPU)
    nmb = args.micro_batch or (4 if pp > 1 else 1)
    zero = args.zero if args.zero is not None else (
        "v1" if args.config == "bert_zero" else "")
    epl.init(epl.Config({
        "pipeline.num_micro_batch": nmb,
        "zero.level": zero,
    }))
    vocab = 30528
    cfg = args.model or "bert-large"
    if args.layers or args.hidden:   # CPU/world-N smoke of this exact path
        h = args.hidden or 256
        cfg = dict(layers=args.layers or 2, hidden=h,
                   heads=max(2, h // 64), ffn=4 * h)
        vocab = 2048
    model = bert.build_bert(cfg, dropout=args.dropout,
                            vocab_size=vocab,
                            max_pos=max(512, args.seq_len), num_stages=pp)
    loss_fn = ParallelCrossEntropy()
    engine = epl.Engine(model, loss_fn=loss_fn, optimizer="adamw", lr=1e-4,
                        dtype=dtype)
    rank = int(os.environ.get("RANK", "0"))
    ids, tgt = bert.synthetic_mlm_batch(batch, args.seq_len, vocab,
                                        device=engine.device,
                                        seed=1234 + rank)
    par = "dp{}".format(engine.num_replicas)
    if pp > 1:
        par += "_pp{}".format(pp)
    if zero:
        par += "_zero_{}".format(zero)
    meta = {"model": (args.model or "bert-large")
            if not (args.layers or args.hidden) else
            "bert-smoke-{}L".format(args.layers or 2),
            "global_batch": batch * engine.num_replicas,
            "seq_len": args.seq_len, "parallelism": par}
    return engine, (ids, tgt), batch * engine.num_replicas, meta


def build_gpt2_bench(args, epl, world, on_gpu, dtype):
    from easyparallellibrary_amd.models import gpt2
    from easyparallellibrary_amd.ops.distributed_losses import (
        ParallelCrossEntropy)
    batch = args.batch or 16
    seq = args.seq_len if args.seq_len != 512 else 1024
    pp = args.pp or (2 if world >= 2 else 1)
    nmb = args.micro_batch or (4 if pp > 1 else 1)
    gc = args.gc if args.gc is not None else "auto"
    offload = args.offload if args.offload is not None else ""
    # world-1 gpt2 steps are launch-heavy (GC recompute doubles the
    # kernel count): hipGraph-capture where safe; world>1 / --dropout /
    # offload fall back to eager inside the engine
    epl.init(epl.Config({
        "pipeline.num_micro_batch": nmb,
        "gradient_checkpoint.type": gc,
        "offload.level": offload,
        "kernel.hip_graph":
            os.environ.get("EPL_KERNEL_HIP_GRAPH", "1") == "1",
    }))
    vocab = 50264
    cfg = args.model or "gpt2-xl"
    if args.layers or args.hidden:   # CPU/world-N smoke of this exact path
        h = args.hidden or 256
        cfg = dict(layers=args.layers or 2, hidden=h,
                   heads=max(2, h // 64), ffn=4 * h)
        vocab = 2048
    model = gpt2.build_gpt2(cfg, dropout=args.dropout,
                            vocab_size=vocab,
                            max_pos=seq, num_stages=pp)
    engine = epl.Engine(model, loss_fn=ParallelCrossEntropy(),
                        optimizer="adamw", lr=1e-4, dtype=dtype)
    rank = int(os.environ.get("RANK", "0"))
    ids, tgt = gpt2.synthetic_lm_batch(batch, seq, vocab,
                                       device=engine.device, seed=99 + rank)
    par = "dp{}".format(engine.num_replicas)
    if pp > 1:
        par += "_pp{}".format(pp)
    if gc:
        par += "_gc"
    if offload:
        par += "_offload"
    meta = {"model": (args.model or "gpt2-xl")
            if not (args.layers or args.hidden) else
            "gpt2-smoke-{}L".format(args.layers or 2),
            "global_batch": batch * engine.num_replicas,
            "seq_len": seq, "parallelism": par}
    return engine, (ids, tgt), batch * engine.num_replicas, meta


def build_resnet_bench(args, epl, world, on_gpu, dtype):
    from easyparallellibrary_amd.models import resnet
    from easyparallellibrary_amd.ops import bridging
    from easyparallellibrary_amd.ops.distributed_losses import (
        ParallelCrossEntropy)
    batch = args.batch or 64
    num_classes = 100000
    epl.init(epl.Config({"cluster.colocate_split_and_replicate": True}))
    model = resnet.build_resnet50_split_classifier(world, num_classes)
    state = {}

    def loss_fn(logits, targets):
        comm = state["engine"].tp_comm
        full_t = bridging.replica_to_split(targets, comm)
        ce = ParallelCrossEntropy(comm=comm,
                                  vocab_begin=state["head"].offset)
        return ce(logits, full_t)

    engine = epl.Engine(model, loss_fn=loss_fn, optimizer="adamw", lr=1e-3,
                        dtype=dtype)
    state["engine"] = engine
    state["head"] = model.head
    rank = int(os.environ.get("RANK", "0"))
    x, y = resnet.synthetic_image_batch(batch, num_classes, size=224,
                                        device=engine.device,
                                        seed=7 + rank)
    x = x.to(dtype)
    meta = {"model": "resnet50-split100k", "global_batch": batch * world,
            "seq_len": 224, "parallelism": "dp{}_tp{}".format(world, world)}
    return engine, (x, y), batch * world, meta


def build_moe_bench(args, epl, world, on_gpu, dtype):
    from easyparallellibrary_amd.models.moe_transformer import (
        build_moe_transformer)
    from easyparallellibrary_amd.models import gpt2
    from easyparallellibrary_amd.ops.distributed_losses import (
        ParallelCrossEntropy)
    batch = args.batch or 8
    seq = args.seq_len if args.seq_len != 512 else 1024
    # the world-1 MoE step is host-launch-bound (~19.6 ms of kernels in
    # a 38 ms step, profiles/r2_final_moe_stats.txt): capture it into a
    # hipGraph (runtime/hipgraph.py).  Ineligible setups (world>1,
    # --dropout) fall back to eager inside the engine.  The DP-scaling
    # configs deliberately stay eager so the driver's N=1 baseline and
    # N>1 ranks run the same step; opt in there with
    # EPL_KERNEL_HIP_GRAPH=1 (bert_dp same-box A/B: 639.7 -> 684.0).
    epl.init(epl.Config({
        "cluster.colocate_split_and_replicate": True,
        "kernel.hip_graph":
            os.environ.get("EPL_KERNEL_HIP_GRAPH", "1") == "1",
    }))
    vocab = 32000
    layers = args.layers or 12
    hidden = args.hidden or 1024
    model = build_moe_transformer(world=world, layers=layers, hidden=hidden,
                                  heads=16, ffn=4 * hidden,
                                  num_experts=max(8, world),
                                  vocab_size=vocab, max_pos=seq)
    engine = epl.Engine(model, loss_fn=ParallelCrossEntropy(),
                        optimizer="adamw", lr=1e-4, dtype=dtype)
    rank = int(os.environ.get("RANK", "0"))
    ids, tgt = gpt2.synthetic_lm_batch(batch, seq, vocab,
                                       device=engine.device, seed=55 + rank)
    meta = {"model": "moe-transformer-{}L".format(layers),
            "global_batch": batch * world,
            "seq_len": seq, "parallelism": "dp{}_ep{}".format(world, world)}
    return engine, (ids, tgt), batch * world, meta


def build_moe_pp_bench(args, epl, world, on_gpu, dtype):
    """PP2 x (DP+EP) MoE hybrid (multi-rank pipeline stages).  ep=2 when
    world divides by 4, else 1; world==1 degrades to the plain MoE
    config."""
    if world < 2:
        return build_moe_bench(args, epl, world, on_gpu, dtype)
    from easyparallellibrary_amd.models.moe_transformer import (
        build_moe_pipeline)
    from easyparallellibrary_amd.models import gpt2
    from easyparallellibrary_amd.ops.distributed_losses import (
        ParallelCrossEntropy)
    batch = args.batch or 8
    seq = args.seq_len if args.seq_len != 512 else 1024
    ep = 2 if world % 4 == 0 else 1
    epl.init(epl.Config({
        "cluster.colocate_split_and_replicate": True,
        "pipeline.num_micro_batch": min(4, batch),
    }))
    vocab = 32000
    layers = args.layers or 12
    hidden = args.hidden or 1024
    model = build_moe_pipeline(stages=2, ep=ep, layers=layers, hidden=hidden,
                               heads=16, ffn=4 * hidden,
                               num_experts=max(8, 2 * ep),
                               vocab_size=vocab, max_pos=seq)
    engine = epl.Engine(model, loss_fn=ParallelCrossEntropy(),
                        optimizer="adamw", lr=1e-4, dtype=dtype)
    rank = int(os.environ.get("RANK", "0"))
    ids, tgt = gpt2.synthetic_lm_batch(batch, seq, vocab,
                                       device=engine.device, seed=55 + rank)
    streams = world // 2  # per-stage positions x replicas
    meta = {"model": "moe-transformer-{}L".format(layers),
            "global_batch": batch * streams,
            "seq_len": seq,
            "parallelism": "pp2_ep{}_dp{}".format(ep, world // (2 * ep))}
    return engine, (ids, tgt), batch * streams, meta


def build_tp_pp_bench(args, epl, world, on_gpu, dtype):
    """PP2 x dense Megatron-TP hybrid (build_tp_pipeline).  tp=2 when
    world divides by 4, else 1; world==1 falls back to bert_dp."""
    if world < 2:
        return build_bert_bench(args, epl, world, on_gpu, dtype)
    from easyparallellibrary_amd.models import gpt2
    from easyparallellibrary_amd.models.tp_transformer import (
        build_tp_pipeline)
    batch = args.batch or 8
    seq = args.seq_len if args.seq_len != 512 else 1024
    tp = 2 if world % 4 == 0 else 1
    epl.init(epl.Config({
        "cluster.colocate_split_and_replicate": True,
        "pipeline.num_micro_batch": min(4, batch),
    }))
    vocab = 32000
    layers = args.layers or 24
    hidden = args.hidden or 1024
    model = build_tp_pipeline(stages=2, tp=tp, layers=layers,
                              hidden=hidden, heads=16, ffn=4 * hidden,
                              vocab_size=vocab, max_pos=seq)

    def lm_loss(logits, targets):
        import torch.nn.functional as F2
        return F2.cross_entropy(logits.reshape(-1, vocab), targets)

    engine = epl.Engine(model, loss_fn=lm_loss, optimizer="adamw",
                        lr=1e-4, dtype=dtype)
    # TP positions shard ONE data stream: seed by replica so every rank
    # of a replica feeds identical tokens (unlike moe_pp, where
    # positions are independent DP+EP streams)
    ids, tgt = gpt2.synthetic_lm_batch(batch, seq, vocab,
                                       device=engine.device,
                                       seed=77 + engine.replica_id)
    streams = world // (2 * tp)
    meta = {"model": "tp-transformer-{}L".format(layers),
            "global_batch": batch * max(1, streams), "seq_len": seq,
            "parallelism": "pp2_tp{}_dp{}".format(tp, max(1, streams))}
    return engine, (ids, tgt.reshape(-1)), batch * max(1, streams), meta


BUILDERS = {
    "bert_dp": build_bert_bench,
    "bert_pp": build_bert_bench,
    "bert_zero": build_bert_bench,
    "gpt2_xl": build_gpt2_bench,
    "resnet_tp": build_resnet_bench,
    "moe": build_moe_bench,
    "moe_pp": build_moe_pp_bench,
    "tp_pp": build_tp_pp_bench,
}


def _enable_tunableop():
    """Load the pre-tuned hipBLASLt algorithm table (profiles/tunableop/)
    so library GEMMs use the algorithms tuned on MI355X — no tuning cost
    at run time.  Opt out with PYTORCH_TUNABLEOP_ENABLED=0."""
    if "PYTORCH_TUNABLEOP_ENABLED" in os.environ:
        return
    here = os.path.dirname(os.path.abspath(__file__))
    base = os.path.join(here, "profiles", "tunableop", "tunableop.csv")
    if os.path.exists(os.path.join(here, "profiles", "tunableop",
                                   "tunableop0.csv")):
        os.environ["PYTORCH_TUNABLEOP_ENABLED"] = "1"
        os.environ["PYTORCH_TUNABLEOP_TUNING"] = "0"
        os.environ["PYTORCH_TUNABLEOP_FILENAME"] = base
        os.environ.setdefault("PYTORCH_TUNABLEOP_VERBOSE", "0")


def main():
    args = parse_args()
    _enable_tunableop()
    import easyparallellibrary_amd as epl

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    on_gpu = torch.cuda.is_available() if args.device != "cpu" else False
    dtype = torch.bfloat16 if on_gpu else torch.float32

    engine, (inputs, targets), samples_per_step, meta = \
        BUILDERS[args.config](args, epl, world, on_gpu, dtype)

    import torch.distributed as dist

    def barrier():
        if dist.is_initialized():
            dist.barrier()
        if on_gpu:
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        engine.train_step(inputs, targets)
    barrier()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        engine.train_step(inputs, targets)
    barrier()
    elapsed = time.perf_counter() - t0

    # MAX over ranks
    if dist.is_initialized():
        et = torch.tensor([elapsed], dtype=torch.float64)
        et_dev = et.to(engine.device) if dist.get_backend() == "nccl" else et
        dist.all_reduce(et_dev, op=dist.ReduceOp.MAX)
        elapsed = float(et_dev.cpu().item())

    value = samples_per_step * args.steps / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    if rank == 0:
        print(json.dumps({
            "metric": "samples_per_sec",
            "value": value,
            "unit": "samples/s",
            "n_gpus": world if on_gpu else world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if dtype == torch.bfloat16 else "fp32",
            "data": "synthetic",
            "config": meta,
        }))


if __name__ == "__main__":
    main()
