"""rocps-launch — single-node cluster spawner.

Launches one process per cluster task on this machine (the reference's
"one command per process" README, automated), pinning each worker to
one GPU of the node:

    python -m rocps.cli.launch --num-ps 1 --num-workers 8 \
        [--issync 1] [--base-port 23400] -- examples/dist_mnist.py \
        --train_steps=200 --batch_size=64

Worker i gets HIP_VISIBLE_DEVICES=i%n_gpus; PS tasks get no GPU.
Exit code is non-zero if any task fails; remaining tasks are terminated
by exact PID (never by pattern).
"""

from __future__ import annotations

import argparse
import os
import signal
import socket
import subprocess
import sys
import time
from typing import List


def _free_ports(n: int) -> List[int]:
    socks, ports = [], []
    for _ in range(n):
        s = socket.socket()
        s.bind(("127.0.0.1", 0))
        socks.append(s)
        ports.append(s.getsockname()[1])
    for s in socks:
        s.close()
    return ports


def main(argv: List[str] = None) -> int:
    ap = argparse.ArgumentParser(prog="rocps-launch")
    ap.add_argument("--num-ps", type=int, default=1)
    ap.add_argument("--num-workers", type=int, default=1)
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--base-port", type=int, default=0,
                    help="0 = pick free ports")
    ap.add_argument("--num-gpus", type=int, default=-1,
                    help="-1 = all visible GPUs")
    ap.add_argument("script")
    ap.add_argument("script_args", nargs=argparse.REMAINDER)
    args = ap.parse_args(argv)

    n = args.num_ps + args.num_workers
    if args.base_port:
        ports = list(range(args.base_port, args.base_port + n))
    else:
        ports = _free_ports(n)
    ps = [f"{args.host}:{p}" for p in ports[:args.num_ps]]
    workers = [f"{args.host}:{p}" for p in ports[args.num_ps:]]

    n_gpus = args.num_gpus
    if n_gpus < 0:
        try:
            import torch
            n_gpus = torch.cuda.device_count()
        except Exception:
            n_gpus = 0

    procs = []
    common = [f"--ps_hosts={','.join(ps)}",
              f"--worker_hosts={','.join(workers)}"]

    def spawn(job: str, idx: int):
        env = dict(os.environ)
        if job == "worker" and n_gpus > 0:
            env["HIP_VISIBLE_DEVICES"] = str(idx % n_gpus)
            env["CUDA_VISIBLE_DEVICES"] = str(idx % n_gpus)
        elif job == "ps":
            env["HIP_VISIBLE_DEVICES"] = ""
            env["CUDA_VISIBLE_DEVICES"] = ""
        cmd = [sys.executable, args.script, *common,
               f"--job_name={job}", f"--task_index={idx}",
               *args.script_args]
        print(f"[rocps-launch] {job}:{idx} -> {' '.join(cmd)}", flush=True)
        p = subprocess.Popen(cmd, env=env)
        procs.append((f"{job}:{idx}", p))

    for i in range(args.num_ps):
        spawn("ps", i)
    for i in range(args.num_workers):
        spawn("worker", i)

    rc = 0
    try:
        pending = dict(procs)
        while pending:
            for name, p in list(pending.items()):
                r = p.poll()
                if r is None:
                    continue
                del pending[name]
                if r != 0:
                    print(f"[rocps-launch] {name} exited rc={r}; "
                          f"terminating cluster", flush=True)
                    if rc == 0:  # first failure wins; peers we then
                        rc = r   # SIGTERM exit -15 and must not mask it
                    for _, q in procs:
                        if q.poll() is None:
                            q.terminate()  # exact PID, never a pattern
            time.sleep(0.2)
    except KeyboardInterrupt:
        for _, q in procs:
            if q.poll() is None:
                q.send_signal(signal.SIGINT)
        rc = 130
    finally:
        deadline = time.time() + 15
        for _, q in procs:
            while q.poll() is None and time.time() < deadline:
                time.sleep(0.1)
            if q.poll() is None:
                q.kill()
    return rc


if __name__ == "__main__":
    sys.exit(main())
