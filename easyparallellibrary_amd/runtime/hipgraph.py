"""hipGraph-captured training step.

MI355X idiom: capture a launch-bound forward+backward into one hipGraph
(`torch.cuda.CUDAGraph` is hipGraph on ROCm) and replay it — one host
submission instead of hundreds of kernel launches.  The reference has no
equivalent (its TF1 runtime amortizes launches in the C++ executor); the
eager PyTorch step pays per-launch host cost instead, and on small-batch
configs that dominates: the MoE bench runs ~19.6 ms of kernels inside a
38 ms step (profiles/r2_final_moe_stats.txt) — the rest is launch gaps.

Scope (checked by :func:`eligible` once, reason logged): the SIMPLE
training path only —

- single-rank world: every communicator is a LocalCommunicator, so the
  gradient-reducer hooks enqueue no collectives and the captured graph
  is pure compute (capturing RCCL ops is deliberately out of scope);
- no AMP loss scaler (bf16 runs scale-free), no ZeRO / CPU offload /
  PBO eager apply / sparse-grad handlers — each of those runs host-side
  logic between backward kernels that a replay would skip.  Gradient
  checkpointing IS allowed: the non-reentrant recompute records its
  kernels inside the captured backward, and dropout-free wrappers skip
  the RNG save/restore;
- no dropout anywhere: torch's graph-safe philox covers torch ops, but
  the in-kernel attention dropout takes its seed as a HOST argument,
  which a replay would freeze;
- micro-batch count 1 and static input shapes (replay asserts them).

The graph captures forward + backward + a loss copy into a static
buffer.  ``Engine.zero_grad`` (arena memset) and the fused optimizer
step stay eager — each is one kernel launch, and keeping them outside
the graph leaves the engine's bookkeeping (found_inf, grad-norm clip,
step counting) untouched.
"""

import logging

import torch
import torch.nn as nn

logger = logging.getLogger(__name__)


def _find_dropout(model):
    """Any active dropout (torch modules or the in-kernel attention p)."""
    for m in model.modules():
        if isinstance(m, nn.Dropout) and m.p > 0:
            return "nn.Dropout(p={})".format(m.p)
        p = getattr(m, "dropout", None)
        if isinstance(p, float) and p > 0:
            return "{}(dropout={})".format(type(m).__name__, p)
    return None


def eligible(engine):
    """(ok, reason) — reason names the first blocking feature."""
    if not torch.cuda.is_available():
        return False, "no GPU"
    if engine.world_size != 1:
        return False, "world_size {} (collectives would be captured)".format(
            engine.world_size)
    if engine.pipeline is not None:
        return False, "pipeline parallelism"
    if engine.num_micro_batch != 1:
        return False, "num_micro_batch {}".format(engine.num_micro_batch)
    if engine.amp.enabled and engine.dtype != torch.bfloat16:
        return False, "AMP loss scaler"
    # gradient checkpointing IS capture-safe here: the non-reentrant
    # recompute runs inside the captured backward and records its
    # kernels, and dropout-free wrappers skip the RNG save/restore
    # (CheckpointWrapper.preserve_rng) — the dropout gate below blocks
    # the only RNG-bearing case
    if engine.config.zero.level:
        return False, "ZeRO"
    if engine.config.offload.level:
        return False, "CPU offload"
    if getattr(engine, "_pbo_eager", False):
        return False, "PreferBackwardOptimizer eager apply"
    if engine.sparse_handlers:
        return False, "sparse-grad handlers"
    drop = _find_dropout(engine._runnable)
    if drop:
        return False, "active dropout ({})".format(drop)
    return True, "ok"


class HipGraphStep:
    """Replaces ``Engine._train_step_simple`` when eligible.

    First ``WARMUP`` calls run the eager step on a side stream (per the
    torch CUDA-graphs recipe: materializes autograd state and hipBLASLt
    workspaces at the capture shapes); the next call captures AND
    replays, then every call replays.  WARMUP=2 keeps the one-time
    capture inside a typical 3-step bench warmup.
    """

    WARMUP = 2

    def __init__(self, engine):
        self.engine = engine
        self.calls = 0
        self.graph = None
        self.static_in = None
        self.static_tgt = None
        self.static_loss = None
        self._warm_stream = torch.cuda.Stream()

    def _warmup_step(self, inputs, targets):
        e = self.engine
        cur = torch.cuda.current_stream()
        self._warm_stream.wait_stream(cur)
        with torch.cuda.stream(self._warm_stream):
            loss = e._train_step_simple(inputs, targets)
        cur.wait_stream(self._warm_stream)
        return loss

    def _capture(self, inputs, targets):
        e = self.engine
        self.static_in = inputs.clone()
        self.static_tgt = targets.clone()
        self.graph = torch.cuda.CUDAGraph()
        e._set_reducers_enabled(True)
        with torch.cuda.graph(self.graph):
            with e.amp.autocast():
                out = e._runnable(self.static_in)
                loss = e.loss_fn(out, self.static_tgt)
            e.amp.scale_loss(loss).backward()
            self.static_loss = loss.detach()
        # hooks that fired DURING capture (python runs once, kernels are
        # only recorded) left the reducers half-advanced; replays never
        # run the hooks, so park the bookkeeping back at step-start
        for r in e.reducers:
            r.reset()
        logger.info("hipGraph captured: input %s, target %s",
                    tuple(self.static_in.shape),
                    tuple(self.static_tgt.shape))

    def run(self, inputs, targets):
        if self.graph is None:
            if self.calls < self.WARMUP:
                self.calls += 1
                return self._warmup_step(inputs, targets)
            try:
                self._capture(inputs, targets)
            except Exception as exc:
                # a model op turned out not to be capture-safe (e.g. an
                # op that syncs): recover the stream, disable graphing
                # for this engine, and run this and all later steps
                # eagerly — nothing executed during the failed capture,
                # so the eager rerun IS the step
                self.graph = None
                torch.cuda.synchronize()
                for r in self.engine.reducers:
                    r.reset()   # hooks may have half-fired mid-capture
                logger.warning(
                    "hipGraph capture failed (%s) — falling back to the "
                    "eager step for this engine", exc)
                self.engine._hipgraph = False
                return self.engine._train_step_simple(inputs, targets)
        if (inputs.shape != self.static_in.shape
                or targets.shape != self.static_tgt.shape):
            raise RuntimeError(
                "hipGraph step captured shapes {}/{} but got {}/{} — "
                "disable kernel.hip_graph for variable-shape batches"
                .format(tuple(self.static_in.shape),
                        tuple(self.static_tgt.shape),
                        tuple(inputs.shape), tuple(targets.shape)))
        self.static_in.copy_(inputs, non_blocking=True)
        self.static_tgt.copy_(targets, non_blocking=True)
        self.graph.replay()
        return self.static_loss
