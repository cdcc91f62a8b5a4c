#!/usr/bin/env python3
"""Benchmark harness (driver contract — see BASELINE.json).

Measures the headline metric: samples/sec (whole node) for BERT-Large
DP(+PP) on synthetic MLM data, bf16, random-init weights.

  python bench.py --gpus N --steps K --warmup W [--model bert-large]
      [--pp S] [--micro-batch M] [--batch B]

For N>1 the driver launches this under torch.distributed.run with one rank
per GPU (RCCL over xGMI); each rank reads RANK/LOCAL_RANK/WORLD_SIZE from
the env.  Weak scaling: per-GPU batch is fixed as N grows.
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
    p.add_argument("--model", default="bert-large")
    p.add_argument("--batch", type=int, default=32,
                   help="per-GPU micro-batch x num_micro_batch samples")
    p.add_argument("--seq-len", type=int, default=512)
    p.add_argument("--pp", type=int, default=1, help="pipeline stages")
    p.add_argument("--micro-batch", type=int, default=1,
                   help="num_micro_batch (pipeline)")
    p.add_argument("--zero", default="", choices=["", "v0", "v1"])
    p.add_argument("--device", default=None)
    return p.parse_args()


def main():
    args = parse_args()
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.models import bert
    from easyparallellibrary_amd.ops.distributed_losses import (
        ParallelCrossEntropy)

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    on_gpu = torch.cuda.is_available() if args.device != "cpu" else False
    dtype = torch.bfloat16 if on_gpu else torch.float32

    cfg = {
        "pipeline.num_micro_batch": args.micro_batch,
        "zero.level": args.zero,
    }
    epl.init(epl.Config(cfg))

    vocab = 30528
    model = bert.build_bert(args.model, vocab_size=vocab,
                            max_pos=max(512, args.seq_len),
                            num_stages=args.pp)
    loss_fn = ParallelCrossEntropy()

    def ce_loss(logits, targets):
        return loss_fn(logits, targets)

    engine = epl.Engine(model, loss_fn=ce_loss, optimizer="adamw", lr=1e-4,
                        dtype=dtype)
    device = engine.device

    ids, targets = bert.synthetic_mlm_batch(
        args.batch, args.seq_len, vocab, device=device, seed=1234 + rank)

    import torch.distributed as dist

    def barrier():
        if dist.is_initialized():
            dist.barrier()
        if on_gpu:
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        engine.train_step(ids, targets)
    barrier()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        engine.train_step(ids, targets)
    barrier()
    elapsed = time.perf_counter() - t0

    # MAX over ranks
    et = torch.tensor([elapsed], dtype=torch.float64)
    if dist.is_initialized():
        et_dev = et.to(device) if dist.get_backend() == "nccl" else et
        dist.all_reduce(et_dev, op=dist.ReduceOp.MAX)
        elapsed = float(et_dev.cpu().item())

    n_gpus = world if on_gpu else world
    # whole-job samples/sec: DP replicas each consume args.batch per step
    num_replicas = engine.num_replicas
    samples_per_step = args.batch * num_replicas
    value = samples_per_step * args.steps / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    if rank == 0:
        par = "dp{}".format(num_replicas)
        if args.pp > 1:
            par += "_pp{}".format(args.pp)
        if args.zero:
            par += "_zero_{}".format(args.zero)
        print(json.dumps({
            "metric": "samples_per_sec",
            "value": value,
            "unit": "samples/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if dtype == torch.bfloat16 else "fp32",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": samples_per_step,
                "seq_len": args.seq_len,
                "parallelism": par,
            },
        }))


if __name__ == "__main__":
    main()
