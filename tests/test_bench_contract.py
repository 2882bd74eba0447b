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
