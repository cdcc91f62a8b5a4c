"""Profiler hooks: step timer, memory timeline CSV, flops report
(reference: profiler/memory_profiler_hook.py, profiler/flops.py)."""

import csv

import torch
import torch.nn as nn

import easyparallellibrary_amd as epl
from easyparallellibrary_amd.profiler.hooks import (FlopsProfiler,
                                                    MemoryProfiler,
                                                    StepTimer)


def _engine():
    epl.init()
    torch.manual_seed(0)
    with epl.replicate(1):
        model = nn.Sequential(nn.Linear(8, 16), nn.Tanh(), nn.Linear(16, 4))
    return epl.Engine(model, loss_fn=nn.MSELoss(), optimizer="adamw",
                      lr=1e-3)


def test_step_timer():
    engine = _engine()
    timer = StepTimer(engine, warmup=1)
    x, y = torch.randn(4, 8), torch.randn(4, 4)
    for _ in range(4):
        loss = timer.step(x, y)
    assert loss is not None
    assert len(timer.times) == 3
    assert timer.mean_ms > 0


def test_memory_profiler_csv(tmp_path):
    prof = MemoryProfiler(every_n_steps=1)
    engine = _engine()
    x, y = torch.randn(4, 8), torch.randn(4, 4)
    for s in range(3):
        engine.train_step(x, y)
        prof.after_step(step=s)
    out = tmp_path / "mem.csv"
    prof.dump_csv(str(out))
    rows = list(csv.DictReader(open(out)))
    assert len(rows) == 3
    assert {"step", "allocated_bytes", "peak_bytes"} <= set(rows[0])
    assert prof.peak_gb >= 0.0


def test_memory_profiler_png(tmp_path):
    prof = MemoryProfiler(every_n_steps=1)
    engine = _engine()
    x, y = torch.randn(4, 8), torch.randn(4, 4)
    for s in range(4):
        engine.train_step(x, y)
        prof.after_step(step=s, note="fwd" if s < 2 else "bwd")
    png = tmp_path / "mem.png"
    ok = prof.dump_png(str(png))
    if ok:  # matplotlib present in this image
        assert png.exists() and png.stat().st_size > 0


def test_flops_profiler_bert_tiny():
    from easyparallellibrary_amd.models import bert
    epl.init()
    model = bert.build_bert(dict(layers=1, hidden=64, heads=2, ffn=128),
                            vocab_size=256, max_pos=32)
    prof = FlopsProfiler(model, seq_len=32, batch=1)
    assert prof.total() > 0
    s = str(prof)
    assert "TOTAL" in s
