"""CPU offload (v0): optimizer states + fp32 master weights live in pinned
host memory; the update step runs on CPU while grads stream D2H and fresh
bf16 weights stream H2D on a dedicated side stream.

Capability parity: /root/reference/epl/parallel/graph_editor.py:727-751
(offload.level='v0': variables + apply ops placed on CPU, weights re-read
lazily) + config epl/config.py:140-145.

MI355X redesign: instead of placing TF variables on CPU, the flat-arena
layout makes offload a buffer-residency choice: the fp32 master arena and
the m/v state arenas are pinned host tensors.  The step is a CHUNKED
SOFTWARE PIPELINE over 64 MB arena slices:
  D2H stream:  grad chunk i+1 (device fp32 convert + hipMemcpyAsync)
  CPU:         torch._fused_adamw_ on chunk i (one multithreaded pass,
               ~7x the eager op chain)
  H2D stream:  updated params of chunk i-1 (direct pinned-fp32 ->
               device-bf16 copy; conversion runs on the GPU)
so copies hide under compute and the optimizer wall-time approaches the
fused-CPU-kernel time alone.  State is 12 bytes/param on host; HBM holds
only bf16 params/grads.
"""

import torch

_CHUNK = 1 << 24  # 16M elements = 64 MB fp32 per pipeline chunk


class CPUOffloadAdamW:
    """AdamW whose state lives on pinned host memory."""

    def __init__(self, groups, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0.01):
        self.groups = list(groups)
        self.lr = lr
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.step_count = 0
        self.pin = torch.cuda.is_available()
        # >64 OpenMP threads OVERSUBSCRIBE the DRAM-bound update on the
        # 256-core GPU hosts (tests/offload_perf_gpu.py: 0.187 s at 128
        # threads vs 0.026 s at 16-64 for a 256M-element fused step)
        if torch.get_num_threads() > 64:
            torch.set_num_threads(64)
        self._d2h = torch.cuda.Stream() if self.pin else None
        self._h2d = torch.cuda.Stream() if self.pin else None
        self._fused = hasattr(torch, "_fused_adamw_")
        for g in self.groups:
            host = lambda: torch.zeros(g.total, dtype=torch.float32,
                                       pin_memory=self.pin)
            g.state["exp_avg"] = host()
            g.state["exp_avg_sq"] = host()
            g.state["master_cpu"] = torch.empty(
                g.total, dtype=torch.float32, pin_memory=self.pin)
            g.state["master_cpu"].copy_(g.master_arena.to("cpu"))
            g.state["grad_cpu"] = torch.empty(
                g.total, dtype=torch.float32, pin_memory=self.pin)
            # free the device-side fp32 master: CPU owns the truth now
            if g.master_arena is not g.param_arena:
                g.master_arena = g.state["master_cpu"]
        # per-bucket D2H overlap state (attach_reducers)
        self._events = {}       # group idx -> [(lo, hi, event)]
        self._n_buckets = {}    # group idx -> bucket count
        # eager mode: a worker thread consumes bucket D2H events DURING
        # backward, runs the CPU AdamW chunk and kicks the H2D of the
        # updated params — by step() time most of the optimizer is done
        self._eager_exec = None
        self._eager_futs = []
        self._eager_active = False
        self._eager_scale = 1.0

    def attach_reducers(self, reducers, groups_of_reducers):
        """Overlap grad D2H with the tail of backward: as each bucket's
        allreduce is enqueued (grads final in the arena on the comm
        stream), copy that span to pinned host memory on the d2h stream
        (NOTES.md round-2 design; closes the ~0.56 s D2H bound of the
        post-backward copy).  Gated to DP reducers without ZeRO shard
        owners or wire compression."""
        if self._d2h is None:
            return 0
        attached = 0
        gi_of = {id(g): i for i, g in enumerate(self.groups)}
        for red, fg in zip(reducers, groups_of_reducers):
            gi = gi_of.get(id(fg))
            if gi is None or red.shard_owners \
                    or not red.supports_bucket_callbacks \
                    or not fg.grad_arena.is_cuda:
                continue
            self._n_buckets[gi] = len(red.buckets)
            red.on_bucket_reduced = self._make_cb(gi)
            attached += 1
        return attached

    def _make_cb(self, gi):
        def cb(bi, lo, hi, comm, owner):
            g = self.groups[gi]
            gc = g.state["grad_cpu"]
            # fence the d2h stream behind BOTH the producing compute
            # stream (grads written during backward; covers the size-1
            # LocalCommunicator whose join is a no-op) and the comm
            # stream (reduced values)
            self._d2h.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(self._d2h):
                comm.join()
                gc[lo:hi].copy_(g.grad_arena[lo:hi].to(torch.float32),
                                non_blocking=True)
                ev = torch.cuda.Event()
                ev.record(self._d2h)
            if self._eager_active:
                self._eager_futs.append(self._eager_exec.submit(
                    self._eager_task, gi, lo, hi, ev))
            else:
                self._events.setdefault(gi, []).append((lo, hi, ev))
        return cb

    # ---- eager CPU apply during backward --------------------------------
    def begin_step(self, grad_scale):
        """Arm the eager pipeline for this step: every reduced bucket's
        CPU AdamW chunk + H2D runs on a worker thread as soon as its
        D2H lands, overlapping the remaining backward.  The engine
        gates this to amp-off / no-grad-clip runs (the scale must be
        final before backward ends).  Safe vs backward param reads for
        the same reason as PreferBackwardOptimizer: a parameter whose
        gradient bucket is complete is not read again this step."""
        import os
        # DEFAULT OFF: same-box A/B measured 931 vs 697 ms/step on the
        # gpt2-xl offload bench — the worker thread contends with the
        # (GC-recompute-heavy, Python-bound) backward for the GIL and
        # LOSES more than the CPU-adamw overlap gains.  Kept behind the
        # flag for non-GC workloads where backward is GIL-light.
        if os.environ.get("EPL_OFFLOAD_EAGER", "0") != "1":
            return False
        if self._d2h is None or not self._n_buckets:
            return False
        if self._eager_exec is None:
            from concurrent.futures import ThreadPoolExecutor
            self._eager_exec = ThreadPoolExecutor(
                max_workers=1, thread_name_prefix="epl-offload")
        self.step_count += 1
        self._eager_scale = float(grad_scale)
        self._eager_futs = []
        self._eager_active = True
        return True

    def _eager_task(self, gi, lo, hi, ev):
        ev.synchronize()   # the bucket's fp32 grads are on the host
        g = self.groups[gi]
        self._chunk_update(g.state["master_cpu"][lo:hi],
                           g.state["grad_cpu"][lo:hi],
                           g.state["exp_avg"][lo:hi],
                           g.state["exp_avg_sq"][lo:hi],
                           self._eager_scale)
        with torch.cuda.stream(self._h2d):
            g.param_arena[lo:hi].copy_(g.state["master_cpu"][lo:hi],
                                       non_blocking=True)

    def _chunk_update(self, w, gc, m, v, grad_scale):
        """One AdamW update on a CPU arena slice."""
        if self._fused:
            torch._fused_adamw_(
                [w], [gc], [m], [v], [],
                [torch.tensor(float(self.step_count))],
                lr=self.lr, beta1=self.beta1, beta2=self.beta2,
                weight_decay=self.weight_decay, eps=self.eps,
                amsgrad=False, maximize=False,
                grad_scale=(torch.tensor(float(grad_scale))
                            if grad_scale != 1.0 else None),
                found_inf=None)
            return
        if grad_scale != 1.0:
            gc.mul_(1.0 / grad_scale)
        m.mul_(self.beta1).add_(gc, alpha=1 - self.beta1)
        v.mul_(self.beta2).addcmul_(gc, gc, value=1 - self.beta2)
        bc1 = 1 - self.beta1 ** self.step_count
        bc2 = 1 - self.beta2 ** self.step_count
        update = (m / bc1) / ((v / bc2).sqrt_().add_(self.eps))
        update.add_(w, alpha=self.weight_decay)
        w.add_(update, alpha=-self.lr)

    def step(self, grad_scale=1.0):
        if self._eager_active:
            self._eager_active = False
            for f in self._eager_futs:
                f.result()   # propagate worker exceptions
            self._eager_futs = []
            self._events.clear()
            # groups whose reducer was not attached (none in practice —
            # attach gates match begin_step's) would be silently
            # skipped; guard loudly instead
            covered = set(self._n_buckets)
            for gi in range(len(self.groups)):
                if gi not in covered:
                    raise RuntimeError(
                        "offload eager step: group {} had no bucket "
                        "callbacks".format(gi))
            torch.cuda.current_stream().wait_stream(self._h2d)
            return
        self.step_count += 1
        for gi, g in enumerate(self.groups):
            gc = g.state["grad_cpu"]
            m, v = g.state["exp_avg"], g.state["exp_avg_sq"]
            w = g.state["master_cpu"]
            spans = [(lo, min(lo + _CHUNK, g.total))
                     for lo in range(0, g.total, _CHUNK)]
            if not g.grad_arena.is_cuda or self._d2h is None:
                for lo, hi in spans:
                    gc[lo:hi].copy_(g.grad_arena[lo:hi].to(torch.float32))
                    self._chunk_update(w[lo:hi], gc[lo:hi], m[lo:hi],
                                       v[lo:hi], grad_scale)
                    g.param_arena[lo:hi].copy_(
                        w[lo:hi].to(g.param_arena.dtype))
                continue
            # D2H: bucket callbacks already copied every span DURING
            # backward (one event per bucket, arena-order) — reuse them;
            # otherwise pre-issue every grad chunk on the copy stream
            # now.  Either way: one event per copied span so the CPU
            # starts as soon as ITS chunk lands.
            bucket_evs = self._events.pop(gi, [])
            if len(bucket_evs) == self._n_buckets.get(gi, -1):
                copies = sorted(bucket_evs)  # (lo, hi, event) by lo
            else:
                self._d2h.wait_stream(torch.cuda.current_stream())
                copies = []
                with torch.cuda.stream(self._d2h):
                    for lo, hi in spans:
                        gc[lo:hi].copy_(
                            g.grad_arena[lo:hi].to(torch.float32),
                            non_blocking=True)
                        ev = torch.cuda.Event()
                        ev.record(self._d2h)
                        copies.append((lo, hi, ev))
            ci = 0
            done = 0   # arena prefix whose events are synchronized
            for lo, hi in spans:
                while done < hi and ci < len(copies):
                    copies[ci][2].synchronize()
                    done = copies[ci][1]
                    ci += 1
                self._chunk_update(w[lo:hi], gc[lo:hi], m[lo:hi],
                                   v[lo:hi], grad_scale)
                # direct pinned-fp32 -> device-bf16 copy: the transfer
                # stays async and the dtype conversion runs on the GPU,
                # keeping the (DRAM-bound) CPU free for the next chunk
                with torch.cuda.stream(self._h2d):
                    g.param_arena[lo:hi].copy_(w[lo:hi],
                                               non_blocking=True)
            torch.cuda.current_stream().wait_stream(self._h2d)

    def zero_grad(self):
        self._events.clear()   # drop stale bucket copies (skipped step)
        for g in self.groups:
            g.zero_grad()

    def state_dict(self):
        return {"step": self.step_count,
                "groups": [{"exp_avg": g.state["exp_avg"],
                            "exp_avg_sq": g.state["exp_avg_sq"],
                            "master": g.state["master_cpu"]}
                           for g in self.groups]}

    def load_state_dict(self, sd):
        self.step_count = sd["step"]
        for g, gs in zip(self.groups, sd["groups"]):
            g.state["exp_avg"].copy_(gs["exp_avg"])
            g.state["exp_avg_sq"].copy_(gs["exp_avg_sq"])
            g.state["master_cpu"].copy_(gs["master"])
            g.param_arena.copy_(
                g.state["master_cpu"].to(g.param_arena.dtype))
