"""GPU async-PS integration (configs[2]-shaped, scaled to one GPU box):
1 host-DRAM PS shard + 2 GPU workers sharing cuda:0, real processes,
shared-memory push/apply/pull. Run with pytest -m gpu."""

import json
import os
import subprocess
import sys

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("needs an MI355X", allow_module_level=True)

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_async_gpu_cluster(tmp_path):
    import socket

    def free(n):
        out = []
        for _ in range(n):
            s = socket.socket()
            s.bind(("127.0.0.1", 0))
            out.append(s.getsockname()[1])
            s.close()
        return out

    ports = free(3)
    ps = [f"127.0.0.1:{ports[0]}"]
    ws = [f"127.0.0.1:{ports[1]}", f"127.0.0.1:{ports[2]}"]
    env = dict(os.environ, ROCPS_BIND_HOST="127.0.0.1")

    def launch(job, i, extra):
        e = dict(env)
        if job == "ps":
            e["HIP_VISIBLE_DEVICES"] = ""
            e["CUDA_VISIBLE_DEVICES"] = ""
        return subprocess.Popen(
            [sys.executable, os.path.join(ROOT, "examples", "dist_mnist.py"),
             f"--ps_hosts={','.join(ps)}", f"--worker_hosts={','.join(ws)}",
             f"--job_name={job}", f"--task_index={i}", "--issync=0",
             "--train_steps=30", "--batch_size=32", *extra],
            cwd=ROOT, env=e, stdout=subprocess.PIPE,
            stderr=subprocess.STDOUT)

    procs = {
        "ps0": launch("ps", 0, []),
        "w0": launch("worker", 0, [f"--loss_log={tmp_path}/l0.json"]),
        "w1": launch("worker", 1, [f"--loss_log={tmp_path}/l1.json"]),
    }
    outs = {}
    for name in ("w0", "w1", "ps0"):
        try:
            out, _ = procs[name].communicate(timeout=240)
        except subprocess.TimeoutExpired:
            procs[name].kill()
            out, _ = procs[name].communicate()
        outs[name] = out.decode(errors="replace")
    for name in ("w0", "w1", "ps0"):
        assert procs[name].returncode == 0, f"{name}:\n{outs[name][-2500:]}"
    l0 = json.load(open(tmp_path / "l0.json"))
    assert len(l0) >= 5 and all(l == l for l in l0)
