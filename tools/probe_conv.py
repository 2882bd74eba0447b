"""Probe MIOpen conv throughput: channels_last vs NCHW, bf16, hot
ResNet-50 shapes (fwd+bwd wall per layer)."""
import sys, os, time
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
torch.backends.cudnn.benchmark = True

def timeit(fn, reps=20, warmup=8):
    for _ in range(warmup): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps * 1e6

shapes = [  # (N, Cin, H, W, Cout, k, stride)
    (256, 3, 224, 224, 64, 7, 2),
    (256, 64, 56, 56, 64, 3, 1),
    (256, 128, 28, 28, 128, 3, 1),
    (256, 256, 14, 14, 256, 3, 1),
    (256, 512, 7, 7, 512, 3, 1),
    (256, 256, 56, 56, 128, 1, 2),
    (256, 1024, 14, 14, 256, 1, 1),
]
dev = "cuda:0"
for fmt_name, fmt in [("NHWC", torch.channels_last), ("NCHW", torch.contiguous_format)]:
    for (N, Ci, H, W, Co, k, s) in shapes:
        conv = torch.nn.Conv2d(Ci, Co, k, stride=s, padding=k // 2, bias=False)
        conv = conv.to(dev).bfloat16().to(memory_format=fmt)
        x = torch.randn(N, Ci, H, W, device=dev).bfloat16().to(memory_format=fmt).requires_grad_(True)
        def fb():
            y = conv(x)
            y.backward(torch.ones_like(y))
            x.grad = None
            conv.weight.grad = None
        us = timeit(fb)
        flops = 2 * N * Co * Ci * k * k * (H // s) * (W // s) * 3  # fwd+dgrad+wgrad
        print(f"{fmt_name} {N}x{Ci}x{H}x{W} -> {Co} k{k}s{s}: {us:8.0f} us  {flops/us/1e6:7.1f} TF/s", flush=True)

# stem variants: Cin=3 is pathological for implicit GEMM — probe padding
print("--- stem Cin pad variants (NHWC) ---", flush=True)
for Ci in (3, 4, 8):
    conv = torch.nn.Conv2d(Ci, 64, 7, stride=2, padding=3, bias=False)
    conv = conv.to(dev).bfloat16().to(memory_format=torch.channels_last)
    x = torch.randn(256, Ci, 224, 224, device=dev).bfloat16().to(
        memory_format=torch.channels_last).requires_grad_(True)
    def fb():
        y = conv(x)
        y.backward(torch.ones_like(y))
        x.grad = None
        conv.weight.grad = None
    us = timeit(fb)
    flops = 2 * 256 * 64 * 3 * 49 * 112 * 112 * 3
    print(f"stem Ci={Ci}: {us:8.0f} us  ({flops/us/1e6:6.1f} TF/s vs Ci=3 work)", flush=True)
