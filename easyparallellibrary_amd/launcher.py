"""Multi-process launcher CLI (``epl-launch`` equivalent).

Capability parity: /root/reference/epl/utils/launcher.py — spawns N
workers (:25-203), free-port selection, per-worker env (TF_CONFIG there;
RANK/WORLD_SIZE/MASTER_* here), GPU visibility partitioning (:118-122),
per-task log files, failure handling (retry :173-188 — here: kill the
job on first worker failure and propagate the exit code; killing is by
exact child PID, never by pattern).

Usage:
    python -m easyparallellibrary_amd.launcher --num_workers 8 \
        [--gpu_per_worker 1] [--log-dir logs/] script.py [script args...]
"""

import argparse
import os
import signal
import socket
import subprocess
import sys
import time


def find_free_port():
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def main(argv=None):
    parser = argparse.ArgumentParser(prog="epl-launch")
    parser.add_argument("--num_workers", type=int, required=True)
    parser.add_argument("--gpu_per_worker", type=int, default=1)
    parser.add_argument("--visible_devices", default=os.environ.get(
        "EPL_CLUSTER_RUN_VISIBLE_DEVICES", ""),
        help="comma-separated physical GPU ids the job may use "
             "(reference cluster.run_visible_devices); workers are "
             "assigned from this list in rank order")
    parser.add_argument("--master_addr", default="127.0.0.1")
    parser.add_argument("--master_port", type=int, default=0)
    parser.add_argument("--log-dir", default="")
    parser.add_argument("script")
    parser.add_argument("script_args", nargs=argparse.REMAINDER)
    args = parser.parse_args(argv)

    port = args.master_port or find_free_port()
    world = args.num_workers
    procs = []
    logs = []
    try:
        for rank in range(world):
            env = dict(os.environ)
            env.update({
                "RANK": str(rank),
                "LOCAL_RANK": str(rank),
                "WORLD_SIZE": str(world),
                "MASTER_ADDR": args.master_addr,
                "MASTER_PORT": str(port),
            })
            if args.gpu_per_worker > 0:
                pool = ([int(d) for d in args.visible_devices.split(",")]
                        if args.visible_devices else None)
                first = rank * args.gpu_per_worker
                ids = range(first, first + args.gpu_per_worker)
                if pool is not None:
                    if first + args.gpu_per_worker > len(pool):
                        raise SystemExit(
                            "visible_devices lists {} GPUs; rank {} needs "
                            "ids {}..{}".format(len(pool), rank, first,
                                                first + args.gpu_per_worker
                                                - 1))
                    ids = [pool[i] for i in ids]
                env["HIP_VISIBLE_DEVICES"] = ",".join(str(i) for i in ids)
                env["LOCAL_RANK"] = "0"
            stdout = None
            if args.log_dir:
                os.makedirs(args.log_dir, exist_ok=True)
                f = open(os.path.join(
                    args.log_dir, "worker_{}.log".format(rank)), "w")
                logs.append(f)
                stdout = f
            procs.append(subprocess.Popen(
                [sys.executable, args.script] + args.script_args,
                env=env, stdout=stdout,
                stderr=subprocess.STDOUT if stdout else None))
        rc = 0
        alive = set(range(world))
        while alive:
            for r in list(alive):
                code = procs[r].poll()
                if code is not None:
                    alive.discard(r)
                    if code != 0:
                        rc = code
                        # kill remaining workers by exact PID
                        for q in procs:
                            if q.poll() is None:
                                q.send_signal(signal.SIGTERM)
            time.sleep(0.2)
        return rc
    finally:
        for q in procs:
            if q.poll() is None:
                q.kill()
        for f in logs:
            f.close()


if __name__ == "__main__":
    sys.exit(main())
