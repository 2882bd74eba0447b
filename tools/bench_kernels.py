"""Per-kernel microbenchmarks on ResNet-50's real shape table (SURVEY
§2d) — µs/call and achieved GB/s vs the torch-native composition.

Run on a GPU box:  python tools/bench_kernels.py [--reps 50]
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch  # noqa: E402


def timeit(fn, reps=50, warmup=10):
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


def line(name, us, nbytes):
    gbs = nbytes / (us * 1e-6) / 1e9
    print(f"{name:58s} {us:9.1f} us  {gbs:8.0f} GB/s", flush=True)


def bench_bn(dev, dtype, reps):
    from rocps.ops import hip_ext
    ext = hip_ext()
    es = torch.tensor([], dtype=dtype).element_size()
    shapes = [(256, 64, 112, 112), (256, 64, 56, 56), (256, 256, 56, 56),
              (256, 128, 28, 28), (256, 512, 28, 28), (256, 256, 14, 14),
              (256, 1024, 14, 14), (256, 512, 7, 7), (256, 2048, 7, 7)]
    for (N, C, H, W) in shapes:
        x = torch.randn(N, C, H, W, device=dev).to(dtype).to(
            memory_format=torch.channels_last)
        dy = torch.randn_like(x)
        w = torch.rand(C, device=dev) + 0.5
        b = torch.randn(C, device=dev) * 0.1
        rm = torch.zeros(C, device=dev)
        rv = torch.ones(C, device=dev)
        nb = x.numel() * es
        tag = f"{N}x{C}x{H}x{W} {str(dtype)[6:]}"

        us = timeit(lambda: ext.bn_relu_fwd_train(x, w, b, rm, rv, 0.1, 1e-5,
                                                  True, None), reps)
        line(f"bn_fwd_train(stats+fin+apply) {tag}", us, 3 * nb)
        y, mean, invstd, fold4c = ext.bn_relu_fwd_train(
            x, w, b, rm, rv, 0.1, 1e-5, True, None)
        us = timeit(lambda: ext.bn_relu_bwd(x, dy, y, w, mean, invstd, True,
                                            False, fold4c), reps)
        line(f"bn_bwd(reduce+fin+apply)      {tag}", us, 7 * nb)

        # torch-native composition for the same op
        x2 = x.detach().clone()
        us = timeit(lambda: torch.relu(torch.nn.functional.batch_norm(
            x2, rm, rv, w, b, True, 0.1, 1e-5)), reps)
        line(f"torch bn+relu fwd             {tag}", us, 3 * nb)


def bench_optim(dev, reps):
    from rocps.ops import fused_optim
    n = 25_600_000
    for dtype in (torch.bfloat16, torch.float32):
        es = torch.tensor([], dtype=dtype).element_size()
        p = torch.randn(n, device=dev).to(dtype)
        g = torch.randn(n, device=dev).to(dtype)
        master = torch.randn(n, device=dev) if dtype == torch.bfloat16 else None
        mom = torch.zeros(n, device=dev)
        nb = n * (2 * es + 8) + (n * 8 if master is not None else 0)
        us = timeit(lambda: fused_optim.momentum(p, g, master, mom, 0.1, 0.9,
                                                 False), reps)
        line(f"fused_momentum 25.6M {str(dtype)[6:]}", us, nb)


def bench_softmax(dev, reps):
    from rocps.ops import hip_ext
    ext = hip_ext()
    for (N, C) in [(256, 1000), (4096, 30522)]:
        for dtype in (torch.bfloat16, torch.float32):
            es = torch.tensor([], dtype=dtype).element_size()
            lg = torch.randn(N, C, device=dev).to(dtype)
            lab = torch.randint(0, C, (N,), device=dev)
            us = timeit(lambda: ext.softmax_xent_fwd(lg, lab), reps)
            line(f"softmax_xent {N}x{C} {str(dtype)[6:]}", us, 4 * N * C * es)


def bench_ln(dev, reps):
    from rocps.ops import hip_ext
    ext = hip_ext()
    for (N, D) in [(32 * 128, 768), (32 * 512, 768)]:
        for dtype in (torch.bfloat16,):
            es = 2
            x = torch.randn(N, D, device=dev).to(dtype)
            w = torch.rand(D, device=dev)
            b = torch.randn(D, device=dev)
            us = timeit(lambda: ext.ln_fwd(x, w, b, 1e-12), reps)
            line(f"ln_fwd {N}x{D} {str(dtype)[6:]}", us, 2 * N * D * es)


def bench_gemm(dev, reps):
    from rocps.ops import hip_ext
    ext = hip_ext()
    for (M, N, K) in [(4096, 4096, 4096), (8192, 768, 768),
                      (8192, 3072, 768), (8192, 768, 3072),
                      (8192, 8192, 8192)]:
        a = (torch.randn(M, K, device=dev) / 8).bfloat16()
        w = (torch.randn(N, K, device=dev) / 8).bfloat16()
        flop = 2.0 * M * N * K
        us = timeit(lambda: ext.gemm_bf16(a, w, None, 0, 1), reps)
        print(f"rocps gemm128  {M}x{N}x{K}: {us:9.1f} us "
              f"{flop / (us * 1e-6) / 1e12:7.1f} TF/s", flush=True)
        if M % 256 == 0 and N % 256 == 0 and K % 128 == 0:
            us = timeit(lambda: ext.gemm_bf16(a, w, None, 0, 2), reps)
            print(f"rocps gemm256  {M}x{N}x{K}: {us:9.1f} us "
                  f"{flop / (us * 1e-6) / 1e12:7.1f} TF/s", flush=True)
            us = timeit(lambda: ext.gemm_bf16(a, w, None, 0, 3), reps)
            print(f"rocps gemm256L {M}x{N}x{K}: {us:9.1f} us "
                  f"{flop / (us * 1e-6) / 1e12:7.1f} TF/s", flush=True)
            us = timeit(lambda: ext.gemm_bf16(a, w, None, 0, 4), reps)
            print(f"rocps gemm256S {M}x{N}x{K}: {us:9.1f} us "
                  f"{flop / (us * 1e-6) / 1e12:7.1f} TF/s", flush=True)
        us = timeit(lambda: a @ w.t(), reps)
        print(f"rocblas matmul {M}x{N}x{K}: {us:9.1f} us "
              f"{flop / (us * 1e-6) / 1e12:7.1f} TF/s", flush=True)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--reps", type=int, default=50)
    ap.add_argument("--what", default="all",
                    choices=["all", "bn", "optim", "softmax", "ln", "gemm"])
    args = ap.parse_args()
    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    if args.what in ("all", "bn"):
        bench_bn(dev, torch.bfloat16, args.reps)
    if args.what in ("all", "optim"):
        bench_optim(dev, args.reps)
    if args.what in ("all", "softmax"):
        bench_softmax(dev, args.reps)
    if args.what in ("all", "ln"):
        bench_ln(dev, args.reps)
    if args.what in ("all", "gemm"):
        bench_gemm(dev, args.reps)


if __name__ == "__main__":
    main()
