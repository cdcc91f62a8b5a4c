"""The driver's exact multi-rank bench launch, on CPU/gloo.

The round-end SCALE run executes
``python -m torch.distributed.run --nnodes=1 --nproc-per-node N
--master-addr 127.0.0.1 --master-port P bench.py --gpus N --steps K
--warmup W`` — this test runs that literal command shape at world 2
with a tiny model so the rendezvous, env parsing, engine build, timed
loop, max-over-ranks reduction and rank-0 JSON contract are covered by
CI (not just by one-off manual smokes)."""

import json
import os
import socket
import subprocess
import sys

import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _launch(extra, nproc=2, timeout=420):
    cmd = [sys.executable, "-m", "torch.distributed.run",
           "--nnodes=1", "--nproc-per-node", str(nproc),
           "--master-addr", "127.0.0.1",
           "--master-port", str(_free_port()),
           os.path.join(REPO, "bench.py"),
           "--gpus", str(nproc), "--steps", "2", "--warmup", "1",
           "--layers", "2", "--hidden", "128", "--device", "cpu"] + extra
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    out = subprocess.run(cmd, capture_output=True, text=True,
                         timeout=timeout, env=env, cwd=REPO)
    assert out.returncode == 0, (out.stdout[-2000:], out.stderr[-2000:])
    line = [l for l in out.stdout.splitlines()
            if l.startswith("{") and '"metric"' in l]
    assert len(line) == 1, out.stdout[-2000:]  # exactly ONE rank prints
    return json.loads(line[0])


def test_driver_scale_launch_world2_default_config():
    rec = _launch([])
    assert rec["n_gpus"] == 2
    assert rec["steps"] == 2 and rec["warmup"] == 1
    assert rec["value"] > 0 and rec["ms_per_step"] > 0
    assert rec["scaling"] == "weak"
    assert rec["config"]["parallelism"] == "dp2"


def test_driver_scale_launch_world2_pp_config():
    rec = _launch(["--config", "bert_pp"])
    assert rec["n_gpus"] == 2
    assert rec["value"] > 0
    assert "pp" in rec["config"]["parallelism"]
