"""conv1x1 GEMM-path vs MIOpen per ResNet-50 shape (ROADMAP §Perf 1).

Run on a GPU box:  python tools/bench_conv1x1.py [--batch 256]
Decides, shape by shape, whether round 2 flips the model dispatch to
rocps.ops.conv1x1 (only where it beats the MIOpen igemm baseline).
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch  # noqa: E402
import torch.nn.functional as F  # noqa: E402

# (Cin, Cout, H, stride) — every distinct 1x1 in ResNet-50 at 224²
SHAPES = [
    (64, 64, 56, 1), (64, 256, 56, 1), (256, 64, 56, 1),
    (256, 128, 56, 1), (256, 512, 56, 2), (128, 512, 28, 1),
    (512, 128, 28, 1), (512, 256, 28, 1), (512, 1024, 28, 2),
    (256, 1024, 14, 1), (1024, 256, 14, 1), (1024, 512, 14, 1),
    (1024, 2048, 14, 2), (512, 2048, 7, 1), (2048, 512, 7, 1),
]


def timeit(fn, reps=30, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(enable_timing=True)
    t1 = torch.cuda.Event(enable_timing=True)
    t0.record()
    for _ in range(reps):
        fn()
    t1.record()
    torch.cuda.synchronize()
    return t0.elapsed_time(t1) * 1e3 / reps  # us


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=256)
    ap.add_argument("--train", action="store_true",
                    help="time fwd+bwd (the dispatch decision metric)")
    args = ap.parse_args()
    torch.backends.cudnn.benchmark = True
    from rocps.ops.conv1x1 import _Conv1x1Fn, conv1x1_nhwc

    winners = []
    mode = "fwd+bwd" if args.train else "fwd"
    print(f"{'shape':>28} {'miopen us':>10} {'gemm us':>10} {'win':>5}"
          f"   ({mode})")
    for cin, cout, h, s in SHAPES:
        x = torch.randn(args.batch, cin, h, h, device="cuda",
                        dtype=torch.bfloat16).contiguous(
            memory_format=torch.channels_last)
        w = torch.randn(cout, cin, 1, 1, device="cuda",
                        dtype=torch.bfloat16)
        if args.train:
            xr = x.requires_grad_(True)
            wr = w.clone().requires_grad_(True)

            def ref_step():
                y = F.conv2d(xr, wr, stride=s)
                y.backward(torch.ones_like(y))
                xr.grad = wr.grad = None

            def gemm_step():
                y = _Conv1x1Fn.apply(xr, wr, s)
                y.backward(torch.ones_like(y))
                xr.grad = wr.grad = None

            t_ref = timeit(ref_step)
            t_gemm = timeit(gemm_step)
        else:
            t_ref = timeit(lambda: F.conv2d(x, w, stride=s))
            t_gemm = timeit(lambda: conv1x1_nhwc(x, w, stride=s))
        flops = 2 * args.batch * (h // s) ** 2 * cin * cout
        if args.train:
            flops *= 3
        win = t_gemm < t_ref
        if win:
            winners.append((cin, cout, s))
        print(f"{(cin, cout, h, s)!s:>28} {t_ref:10.1f} {t_gemm:10.1f} "
              f"{'GEMM' if win else 'ref':>5}"
              f"   ({flops / t_gemm / 1e6:.0f} TF/s gemm)")
    print("dispatch set (paste into rocps/ops/conv1x1.py _GEMM_SHAPES):")
    print(sorted(set(winners)))


if __name__ == "__main__":
    if "--wgrad" not in sys.argv:
        main()


def bench_wgrad(batch=256):
    """A/B the hand-written wgrad kernel vs MIOpen's wrw per shape."""
    import torch
    import torch.nn.functional as F

    from rocps.ops import hip_ext
    ext = hip_ext()
    print(f"{'shape':>26} {'miopen us':>10} {'custom us':>10} {'win':>5}")
    for cin, cout, h, s in SHAPES:
        if s != 1:
            continue
        m = batch * h * h
        x = torch.randn(batch, cin, h, h, device="cuda",
                        dtype=torch.bfloat16).contiguous(
            memory_format=torch.channels_last)
        dy = torch.randn(batch, cout, h, h, device="cuda",
                         dtype=torch.bfloat16).contiguous(
            memory_format=torch.channels_last)
        w = torch.randn(cout, cin, 1, 1, device="cuda",
                        dtype=torch.bfloat16)
        xr = x.permute(0, 2, 3, 1).reshape(m, cin)
        dyr = dy.permute(0, 2, 3, 1).reshape(m, cout)

        def miopen_wrw():
            torch.ops.aten.convolution_backward(
                dy, x, w, None, [1, 1], [0, 0], [1, 1], False, [0, 0], 1,
                [False, True, False])

        def custom():
            ext.conv1x1_wgrad(dyr, xr)

        t_ref = timeit(miopen_wrw)
        t_cus = timeit(custom)
        print(f"{(cin, cout, h)!s:>26} {t_ref:10.1f} {t_cus:10.1f} "
              f"{'CUST' if t_cus < t_ref else 'ref':>5}")


if __name__ == "__main__" and "--wgrad" in sys.argv:
    torch.backends.cudnn.benchmark = True
    bench_wgrad()
