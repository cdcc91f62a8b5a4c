"""Runtime profiler hooks.

Capability parity: /root/reference/epl/profiler/memory_profiler_hook.py
(MemoryProfilerHook: per-step memory timeline -> CSV, :32-271) and
flops.py (FlopsProfilerHook one-shot report :133-158).

MI355X redesign: torch.cuda memory stats + wall-clock step timing instead
of TF RunMetadata; deeper kernel-level profiling is rocprofv3's job (the
bench harness records those separately under profiles/).
"""

import csv
import time

import torch


class StepTimer:
    """Rolling step-time stats; wraps engine.train_step."""

    def __init__(self, engine, warmup=3):
        self.engine = engine
        self.warmup = warmup
        self.times = []
        self._step = 0

    def step(self, *args, **kwargs):
        if self.engine.device.type == "cuda":
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        out = self.engine.train_step(*args, **kwargs)
        if self.engine.device.type == "cuda":
            torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        self._step += 1
        if self._step > self.warmup:
            self.times.append(dt)
        return out

    @property
    def mean_ms(self):
        if not self.times:
            return float("nan")
        return sum(self.times) / len(self.times) * 1000.0


class MemoryProfiler:
    """Record device-memory stats every N steps; dump_csv writes the
    timeline (reference: MemoryProfilerHook CSV output)."""

    def __init__(self, every_n_steps=1):
        self.every = every_n_steps
        self.records = []
        self._step = 0

    def after_step(self, step=None, note=""):
        self._step += 1
        if self._step % self.every:
            return
        if torch.cuda.is_available():
            alloc = torch.cuda.memory_allocated()
            peak = torch.cuda.max_memory_allocated()
            reserved = torch.cuda.memory_reserved()
        else:
            alloc = peak = reserved = 0
        self.records.append({
            "step": step if step is not None else self._step,
            "allocated_bytes": alloc,
            "peak_bytes": peak,
            "reserved_bytes": reserved,
            "note": note,
        })

    def dump_csv(self, path):
        if not self.records:
            return
        with open(path, "w", newline="") as f:
            w = csv.DictWriter(f, fieldnames=list(self.records[0]))
            w.writeheader()
            w.writerows(self.records)

    def dump_png(self, path):
        """Memory timeline plot (reference: MemoryProfilerHook's
        matplotlib PNG with per-phase coloring, memory_profiler_hook.py
        :207-271 — notes recorded per step play the phase role here).
        No-op when matplotlib is unavailable."""
        if not self.records:
            return False
        try:
            import matplotlib
            matplotlib.use("Agg")
            import matplotlib.pyplot as plt
        except ImportError:
            return False
        steps = [r["step"] for r in self.records]
        gib = 1 << 30
        fig, ax = plt.subplots(figsize=(8, 4))
        ax.plot(steps, [r["allocated_bytes"] / gib for r in self.records],
                label="allocated")
        ax.plot(steps, [r["peak_bytes"] / gib for r in self.records],
                label="peak", linestyle="--")
        ax.plot(steps, [r["reserved_bytes"] / gib for r in self.records],
                label="reserved", linestyle=":")
        # color the background by note runs (the phase coloring analogue)
        notes = [r["note"] for r in self.records]
        colors = {}
        palette = ["#fde2cf", "#d4e7fa", "#d9f2d9", "#f5d9f0", "#f7f6cf"]
        start = 0
        for i in range(1, len(notes) + 1):
            if i == len(notes) or notes[i] != notes[start]:
                note = notes[start]
                if note:
                    c = colors.setdefault(
                        note, palette[len(colors) % len(palette)])
                    ax.axvspan(steps[start], steps[i - 1], color=c,
                               alpha=0.5,
                               label=note if note not in ax.get_legend_handles_labels()[1] else None)
                start = i
        ax.set_xlabel("step")
        ax.set_ylabel("GiB")
        ax.set_title("device memory timeline")
        ax.legend(loc="upper left", fontsize=8)
        fig.tight_layout()
        fig.savefig(path)
        plt.close(fig)
        return True

    @property
    def peak_gb(self):
        if not self.records:
            return 0.0
        return max(r["peak_bytes"] for r in self.records) / (1 << 30)


class FlopsProfiler:
    """One-shot model flops report (reference: FlopsProfilerHook)."""

    def __init__(self, model, seq_len=512, batch=1):
        from easyparallellibrary_amd.profiler.cost_model import profile_flops
        self.report = profile_flops(model, seq_len, batch)

    def total(self):
        return sum(self.report.values())

    def __str__(self):
        lines = ["flops per sample (fwd):"]
        for k, v in self.report.items():
            lines.append("  {:30s} {:>14,d}".format(k, int(v)))
        lines.append("  {:30s} {:>14,d}".format("TOTAL", int(self.total())))
        return "\n".join(lines)
