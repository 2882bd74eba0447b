"""rocps-launch end-to-end tests: the single-node
cluster spawner drives examples/dist_mnist.py through the reference's
per-process CLI, and tears the whole cluster down if any task fails.
"""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _launchpy(*argv, timeout):
    env = dict(os.environ)
    env.setdefault("ROCPS_TERMINATE_BT", "1")
    return subprocess.run(
        [sys.executable, "-m", "rocps.cli.launch", *argv],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=timeout)


def test_launch_sync_cluster_end_to_end(tmp_path):
    log_tpl = str(tmp_path / "w{task}.json")  # fans out per worker
    r = _launchpy(
        "--num-ps", "1", "--num-workers", "2", "--num-gpus", "0", "--",
        "examples/dist_mnist.py", "--issync=1", "--train_steps=8",
        "--batch_size=8", f"--loss_log={log_tpl}", timeout=240)
    assert r.returncode == 0, f"stdout:\n{r.stdout}\nstderr:\n{r.stderr}"
    assert "[rocps-launch] ps:0" in r.stdout
    assert "[rocps-launch] worker:1" in r.stdout
    for w in (0, 1):
        losses = json.load(open(str(tmp_path / f"w{w}.json")))
        assert len(losses) >= 8


def test_launch_dist_train_zoo_script(tmp_path):
    """examples/dist_train.py (the model-zoo variant of the reference
    script) end-to-end on CPU: mnist_mlp, sync, 2 workers, with JSONL
    metrics + TB event files in --logdir."""
    logdir = str(tmp_path / "logs")
    r = _launchpy(
        "--num-ps", "1", "--num-workers", "2", "--num-gpus", "0", "--",
        "examples/dist_train.py", "--model=mnist_mlp", "--issync=1",
        "--train_steps=6", "--batch_size=8", "--dtype=fp32",
        f"--logdir={logdir}", timeout=240)
    assert r.returncode == 0, f"stdout:\n{r.stdout}\nstderr:\n{r.stderr}"
    files = os.listdir(logdir)
    assert "metrics.jsonl" in files
    assert any(f.startswith("events.out.tfevents.") for f in files)


def test_ps_server_cli(tmp_path):
    """rocps.cli.ps_server (the standalone PS task bench.py spawns):
    serve a shard to an in-process worker engine, then shut down."""
    import socket

    import torch

    from rocps.cluster import ClusterSpec, ControlClient, Server, \
        shutdown_cluster
    from rocps.engine.async_engine import AsyncWorkerEngine
    from rocps.models import mnist_mlp, synthetic_mnist_batch
    from rocps.train.optimizer import GradientDescentOptimizer

    socks, ports = [], []
    for _ in range(2):
        s = socket.socket()
        s.bind(("127.0.0.1", 0))
        socks.append(s)
        ports.append(s.getsockname()[1])
    for s in socks:
        s.close()
    spec = {"ps": [f"127.0.0.1:{ports[0]}"],
            "worker": [f"127.0.0.1:{ports[1]}"]}
    cluster = ClusterSpec(spec)
    # spawn the ps FIRST: it hosts the cluster master (kv), and a
    # starting Server retries registration against it
    ps = subprocess.Popen(
        [sys.executable, "-m", "rocps.cli.ps_server",
         "--cluster-json", json.dumps(spec), "--task-index", "0"],
        cwd=REPO, env=dict(os.environ, ROCPS_BIND_HOST="127.0.0.1"),
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT)
    worker_server = Server(cluster, "worker", 0)
    try:
        torch.manual_seed(5)
        m = mnist_mlp(hidden=16)
        eng = AsyncWorkerEngine(m, cluster,
                                ControlClient(cluster.master_address),
                                task_index=0, device=torch.device("cpu"))
        opt = GradientDescentOptimizer(0.5)
        eng.initialize(opt)
        g = torch.Generator().manual_seed(1)
        x, y = synthetic_mnist_batch(32, generator=g)
        first = last = None
        for _ in range(8):
            eng.zero_grad()
            loss = m.loss(x, y)
            eng.step(loss, opt)
            last = float(loss.detach())
            first = first if first is not None else last
        assert last < first
        eng.detach()
        shutdown_cluster(cluster)
        out, _ = ps.communicate(timeout=60)
        assert ps.returncode == 0, out.decode(errors="replace")
    finally:
        if ps.poll() is None:
            ps.kill()
        worker_server.stop()


def test_launch_propagates_task_failure(tmp_path):
    bad = tmp_path / "bad_task.py"
    bad.write_text(
        "import sys\n"
        "# fail only in worker role; ps would otherwise serve forever\n"
        "if any(a.startswith('--job_name=worker') for a in sys.argv):\n"
        "    sys.exit(3)\n"
        "import time\n"
        "time.sleep(60)\n")
    r = _launchpy("--num-ps", "1", "--num-workers", "1", "--num-gpus", "0",
                  "--", str(bad), timeout=120)
    assert r.returncode == 3, f"stdout:\n{r.stdout}\nstderr:\n{r.stderr}"
    assert "terminating cluster" in r.stdout
