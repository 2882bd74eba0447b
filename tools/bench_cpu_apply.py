"""Fused CPU PS-apply vs the torch op-at-a-time path (ROADMAP §Perf 4).

Runs anywhere (no GPU):  python tools/bench_cpu_apply.py [--numel N]
Reported in profiles/RESULTS.md; the bf16+master rows are the actual
async-PS configuration on a GPU box.
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch  # noqa: E402


def bench(opt_name, fused, numel, gdtype, pdtype, reps=20):
    os.environ["ROCPS_CPU_FUSED"] = "1" if fused else "0"
    import importlib

    import rocps.train.optimizer as O
    importlib.reload(O)
    if opt_name == "sgd":
        opt = O.GradientDescentOptimizer(0.1)
    elif opt_name == "momentum":
        opt = O.MomentumOptimizer(0.1, 0.9)
    else:
        opt = O.AdamOptimizer(0.01)
        opt._t = 5
    p = torch.randn(numel).to(pdtype)
    g = torch.randn(numel).to(gdtype)
    opt.apply_flat(p, g, slot=0)
    t0 = time.perf_counter()
    for _ in range(reps):
        opt.apply_flat(p, g, slot=0)
    return (time.perf_counter() - t0) / reps


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--numel", type=int, default=12_800_000)
    args = ap.parse_args()
    print(f"shard numel = {args.numel/1e6:.1f}M, "
          f"torch threads = {torch.get_num_threads()}")
    print(f"{'case':>40} {'torch ms':>9} {'fused ms':>9} {'speedup':>8}")
    cases = [
        ("sgd", torch.float32, torch.float32),
        ("momentum", torch.float32, torch.float32),
        ("momentum", torch.bfloat16, torch.bfloat16),
        ("momentum", torch.bfloat16, torch.float32),
        ("adam", torch.float32, torch.float32),
        ("adam", torch.bfloat16, torch.bfloat16),
    ]
    for name, gd, pd in cases:
        tt = bench(name, False, args.numel, gd, pd)
        tf = bench(name, True, args.numel, gd, pd)
        label = f"{name} g={str(gd)[6:]} p={str(pd)[6:]}"
        print(f"{label:>40} {tt*1e3:9.2f} {tf*1e3:9.2f} {tt/tf:7.2f}x")


if __name__ == "__main__":
    main()
