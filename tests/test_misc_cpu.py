"""IO slicing, metric merging, launcher, AMP scaler units."""

import os
import subprocess
import sys

import pytest
import torch

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

from easyparallellibrary_amd.utils.io_slicing import (slice_dataset_indices,
                                                      slice_files)


def test_slice_files_balanced():
    files = [str(i) for i in range(10)]
    parts = [slice_files(files, 4, r) for r in range(4)]
    assert [len(p) for p in parts] == [3, 3, 2, 2]
    assert sum(parts, []) == files


def test_slice_indices():
    idx = [list(slice_dataset_indices(10, 3, r)) for r in range(3)]
    assert sum(idx, []) == list(range(10))


def test_dynamic_loss_scaler():
    from easyparallellibrary_amd.runtime.amp import DynamicLossScaler
    s = DynamicLossScaler(init_scale=1024, growth_interval=2)
    s.update(False)
    s.update(False)
    assert s.scale == 2048
    s.update(True)
    assert s.scale == 1024
    s.update(True)
    assert s.scale == 512


def test_launcher_cli():
    """2-worker launcher runs a trivial script and propagates env."""
    script = (
        "import os,sys;"
        "assert os.environ['WORLD_SIZE']=='2';"
        "print('rank', os.environ['RANK'])"
    )
    import tempfile
    with tempfile.NamedTemporaryFile("w", suffix=".py", delete=False) as f:
        f.write(script)
        path = f.name
    rc = subprocess.call(
        [sys.executable, "-m", "easyparallellibrary_amd.launcher",
         "--num_workers", "2", "--gpu_per_worker", "0", path])
    assert rc == 0


def test_launcher_failure_propagates():
    import tempfile
    with tempfile.NamedTemporaryFile("w", suffix=".py", delete=False) as f:
        f.write("import sys; sys.exit(3)")
        path = f.name
    rc = subprocess.call(
        [sys.executable, "-m", "easyparallellibrary_amd.launcher",
         "--num_workers", "2", "--gpu_per_worker", "0", path])
    assert rc != 0


def test_metric_merge_single():
    import easyparallellibrary_amd as epl
    import torch.nn as nn
    epl.init()
    with epl.replicate(1):
        m = nn.Linear(2, 2)
    engine = epl.Engine(m, loss_fn=nn.MSELoss())
    out = engine.all_reduce_metric(3.0)
    assert float(out) == 3.0
    files = engine.slice_input_files(["a", "b"])
    assert files == ["a", "b"]


def test_collections_api():
    import easyparallellibrary_amd as epl
    epl.init()
    epl.add_to_collection(1.5, epl.GraphKeys.GLOBAL_MEAN_OBJECTS)
    epl.add_to_collection(2.5, epl.GraphKeys.GLOBAL_MEAN_OBJECTS)
    assert epl.get_collection(epl.GraphKeys.GLOBAL_MEAN_OBJECTS) == [1.5, 2.5]

    import torch.nn as nn
    with epl.replicate(1):
        m = nn.Linear(2, 2)
    engine = epl.Engine(m, loss_fn=nn.MSELoss())
    merged = engine.merged_collections()
    vals = [float(v) for v in merged[epl.GraphKeys.GLOBAL_MEAN_OBJECTS]]
    assert vals == [1.5, 2.5]


def test_grouped_apply_config():
    import torch
    import torch.nn as nn
    import easyparallellibrary_amd as epl
    epl.init(epl.Config({"optimizer.num_apply_group": 3}))
    torch.manual_seed(9)
    with epl.replicate(1):
        m = nn.Linear(8, 8)
    engine = epl.Engine(m, loss_fn=nn.MSELoss(), lr=1e-2)
    x = torch.randn(4, 8)
    y = torch.randn(4, 8)
    l0 = float(engine.train_step(x, y))
    l1 = float(engine.train_step(x, y))
    assert l1 < l0


def test_launcher_runs_dp_training():
    """End-to-end: epl-launch spawns 2 workers that train a DP model."""
    import os
    import tempfile
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    script = """
import sys
sys.path.insert(0, {!r})
import torch, torch.nn as nn
import easyparallellibrary_amd as epl""".format(repo) + """
epl.init()
torch.manual_seed(1)
with epl.replicate(1):
    m = nn.Linear(4, 2)
engine = epl.Engine(m, loss_fn=nn.MSELoss(), lr=1e-2)
x = torch.randn(4, 4); y = torch.randn(4, 2)
l0 = float(engine.train_step(x, y))
l1 = float(engine.train_step(x, y))
assert l1 < l0, (l0, l1)
assert engine.world_size == 2
print("worker ok", engine.rank)
"""
    with tempfile.NamedTemporaryFile("w", suffix=".py", delete=False) as f:
        f.write(script)
        path = f.name
    rc = subprocess.call(
        [sys.executable, "-m", "easyparallellibrary_amd.launcher",
         "--num_workers", "2", "--gpu_per_worker", "0", path])
    assert rc == 0


@pytest.mark.parametrize("script", ["train_bert_dp.py",
                                    "train_bert_pipeline.py",
                                    "train_bert_zero.py",
                                    "train_moe.py",
                                    "train_long_context.py",
                                    "train_tp_pipeline.py"])
def test_examples_tiny_cpu(script):
    """Every example runs end-to-end in tiny mode on CPU."""
    import subprocess
    import sys
    env = dict(os.environ, EPL_EXAMPLE_TINY="1")
    r = subprocess.run(
        [sys.executable, os.path.join(REPO_ROOT, "examples", script)],
        env=env, capture_output=True, text=True, timeout=240)
    assert r.returncode == 0, r.stderr[-2000:]
    assert "step 9" in r.stdout, r.stdout[-500:]


def test_io_slicing_tiling_randomized():
    import random
    from easyparallellibrary_amd.utils.io_slicing import slice_files
    rng = random.Random(8)
    for _ in range(100):
        w = rng.randint(1, 8)
        n = rng.randint(w, 500)
        files = [str(i) for i in range(n)]
        for unbalanced in (False, True):
            parts = [slice_files(files, w, r, unbalanced=unbalanced)
                     for r in range(w)]
            assert sum(parts, []) == files
        dropped = [slice_files(files, w, r, drop_last=True)
                   for r in range(w)]
        sizes = {len(p) for p in dropped}
        assert len(sizes) == 1
        assert sum(dropped, []) == files[:n - n % w]
