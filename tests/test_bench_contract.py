"""bench.py is a driver contract: one JSON line on stdout with the
agreed fields. Exercise it on CPU with the MNIST config."""

import json
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract():
    out = subprocess.run(
        [sys.executable, os.path.join(ROOT, "bench.py"), "--model",
         "mnist_mlp", "--steps", "3", "--warmup", "1", "--batch", "16"],
        cwd=ROOT, capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines()
             if l.startswith("{")]
    assert len(lines) == 1, out.stdout
    rec = json.loads(lines[0])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in rec, key
    assert rec["n_gpus"] == 1 and rec["steps"] == 3 and rec["warmup"] == 1
    assert rec["data"] == "synthetic"
    assert rec["value"] > 0
    assert rec["config"]["global_batch"] == 16
    assert rec["config"]["parallelism"] == "dp1"


def test_bench_torchrun_two_ranks_contract():
    """The driver's exact multi-GPU launch shape, on CPU/gloo: one
    process per rank via torch.distributed.run, rank 0 prints ONE JSON
    line whose value is the WHOLE-JOB aggregate."""
    import socket

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    env = dict(os.environ, MASTER_ADDR="127.0.0.1",
               GLOO_SOCKET_IFNAME="lo")
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port), os.path.join(ROOT, "bench.py"),
         "--gpus", "2", "--model", "mnist_mlp", "--steps", "3",
         "--warmup", "1", "--batch", "16"],
        cwd=ROOT, env=env, capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-3000:]
    lines = [l for l in out.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout   # only rank 0 prints
    rec = json.loads(lines[0])
    assert rec["n_gpus"] == 2
    assert rec["config"]["parallelism"] == "dp2"
    # whole-job aggregate: global batch = per-rank byte 16 x 2 ranks
    assert rec["config"]["global_batch"] == 32
    assert rec["value"] > 0


def test_bench_async_mode_contract():
    """bench.py --mode async spawns PS subprocesses and prints the same
    JSON contract (CPU: shm push/apply/pull with the fused CPU apply).
    One internal retry with a fresh port block — the PS/control ports
    are derived from MASTER_PORT and can collide with a previous
    test's lingering sockets."""
    import socket

    last = None
    for attempt in range(2):
        s = socket.socket()
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
        s.close()
        env = dict(os.environ, MASTER_PORT=str(port),
                   MASTER_ADDR="127.0.0.1")
        try:
            out = subprocess.run(
                [sys.executable, os.path.join(ROOT, "bench.py"), "--model",
                 "mnist_mlp", "--mode", "async", "--num-ps", "2", "--steps",
                 "3", "--warmup", "1", "--batch", "16"],
                cwd=ROOT, env=env, capture_output=True, text=True,
                timeout=120)
            assert out.returncode == 0, out.stderr[-3000:]
            lines = [l for l in out.stdout.strip().splitlines()
                     if l.startswith("{")]
            assert len(lines) == 1, out.stdout
            rec = json.loads(lines[0])
            assert rec["config"]["mode"] == "async"
            assert rec["config"]["num_ps"] == 2
            assert rec["value"] > 0
            return
        except (AssertionError, subprocess.TimeoutExpired) as e:
            last = e
    raise last
