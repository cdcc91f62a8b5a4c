"""Multi-process test helper: spawn N ranks over gloo on 127.0.0.1
(mirrors the reference's mocked-cluster tests)."""

import os
import pickle
import tempfile

import torch.multiprocessing as mp


def _worker(rank, world, port, fn, args, result_dir):
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    try:
        out = fn(rank, world, *args)
        with open(os.path.join(result_dir, "r{}.pkl".format(rank)), "wb") as f:
            pickle.dump(("ok", out), f)
    except Exception as e:  # noqa: BLE001
        import traceback
        with open(os.path.join(result_dir, "r{}.pkl".format(rank)), "wb") as f:
            pickle.dump(("err", "{}\n{}".format(e, traceback.format_exc())), f)
        raise


def run_multiprocess(fn, world=2, args=(), timeout=180):
    """Run fn(rank, world, *args) in `world` processes; returns list of
    results ordered by rank; raises on any failure."""
    import random
    port = random.randint(20000, 50000)
    with tempfile.TemporaryDirectory() as d:
        ctx = mp.get_context("spawn")
        procs = []
        for r in range(world):
            p = ctx.Process(target=_worker,
                            args=(r, world, port, fn, args, d))
            p.start()
            procs.append(p)
        for p in procs:
            p.join(timeout)
            if p.is_alive():
                for q in procs:
                    q.terminate()
                raise TimeoutError("multiprocess test timed out")
        results = []
        for r in range(world):
            path = os.path.join(d, "r{}.pkl".format(r))
            if not os.path.exists(path):
                raise RuntimeError("rank {} produced no result".format(r))
            with open(path, "rb") as f:
                status, out = pickle.load(f)
            if status != "ok":
                raise RuntimeError("rank {} failed:\n{}".format(r, out))
            results.append(out)
        return results
