"""1x1 convolution as an MFMA GEMM — round-2 staging of ROADMAP §Perf 1.

A stride-1 1x1 conv over an NHWC (channels_last) tensor IS the GEMM
``rows(N*H*W, Cin) @ W(Cout, Cin)^T`` — exactly ``matmul_nt``'s NT
contract — and 1x1 convs carry ~60% of ResNet-50's conv FLOPs. This
module provides the shape/stride/padding plumbing, CPU-verified against
``torch.nn.functional.conv2d``:

- stride s > 1: spatial subsample (the 1x1 kernel never overlaps), then
  the same GEMM over the surviving positions;
- the MFMA kernel's tile contract (M%128 == N%128 == 0, K%64 == 0) is
  met by zero-padding M (position rows) and N (output channels) up to
  the tile, slicing the result back. K == Cin must already be %64 —
  true for every 1x1 in ResNet-50/WRN past the stem.

``forward_supported`` reports whether the GEMM path applies without
K-padding (which would change the reduction and is not offered).

On CPU (or when the extension is absent) the same plumbing runs over a
plain fp32 matmul, which is what the parity tests pin down.  The
``Conv1x1`` module below carries the measured per-shape dispatch: as
of round 2 the MFMA GEMM (fwd + dgrad) with MIOpen's wrw kernel for
the weight gradient wins 10 of 15 ResNet-50 1x1 shapes
[profiles/conv1x1_train_r2b.txt].
"""

from __future__ import annotations

import torch

from rocps.ops import hip_available

__all__ = ["conv1x1_nhwc", "forward_supported"]

_TM, _TN, _TK = 128, 128, 64


def forward_supported(cin: int) -> bool:
    """K (=Cin) cannot be padded without changing the reduction."""
    return cin % _TK == 0


def _pad_to(x: torch.Tensor, dim: int, mult: int) -> torch.Tensor:
    rem = x.shape[dim] % mult
    if rem == 0:
        return x
    pad_shape = list(x.shape)
    pad_shape[dim] = mult - rem
    return torch.cat([x, x.new_zeros(pad_shape)], dim=dim)


def conv1x1_nhwc(x: torch.Tensor, weight: torch.Tensor,
                 stride: int = 1) -> torch.Tensor:
    """x (N,C,H,W) channels_last, weight (Cout,Cin,1,1) -> NHWC output.

    GEMM path on GPU bf16 (MFMA kernel), reference matmul elsewhere.
    """
    n, cin, h, w = x.shape
    cout = weight.shape[0]
    if not forward_supported(cin):
        raise ValueError(f"Cin={cin} not a multiple of {_TK}")
    if stride > 1:
        x = x[:, :, ::stride, ::stride]
        h, w = x.shape[2], x.shape[3]
    # channels_last rows are (N*H*W, C)
    rows = (x.contiguous(memory_format=torch.channels_last)
            .permute(0, 2, 3, 1).reshape(n * h * w, cin))
    wmat = weight.reshape(cout, cin)
    m = rows.shape[0]
    if (hip_available() and rows.is_cuda and rows.dtype == torch.bfloat16):
        from rocps.ops.gemm import matmul_nt
        rows_p = _pad_to(rows, 0, _TM)
        w_p = _pad_to(wmat, 0, _TN)
        out = matmul_nt(rows_p, w_p)[:m, :cout]
    else:
        out = rows.float() @ wmat.float().t()
        out = out.to(x.dtype)
    return (out.reshape(n, h, w, cout).permute(0, 3, 1, 2)
            .contiguous(memory_format=torch.channels_last))


# ---------------------------------------------------------------------------
# Training path: Conv1x1 module with autograd (round-2 dispatch)
# ---------------------------------------------------------------------------
#
# forward : out_rows = rows(N*Ho*Wo, Cin) @ W(Cout, Cin)^T     [MFMA GEMM]
# dgrad   : dx_rows  = dy_rows(M, Cout) @ (W^T)(Cin, Cout)^T   [MFMA GEMM]
#           (the weight transpose is a (Cout x Cin) copy — trivially small
#           next to the activation tensors)
# wgrad   : MIOpen's wrw asm kernel on GPU (convolution_backward with
#           the weight-only output mask — the library kernel for the
#           tall-skinny sum over M, exactly like rocBLAS for plain
#           GEMMs); rocBLAS/fp32 TN on CPU
#
# Dispatch is PER SHAPE: only (cin, cout, stride) combinations measured
# faster than MIOpen's igemm asm (tools/bench_conv1x1.py, fwd+bwd) run
# the GEMM path; everything else stays on F.conv2d.  ROCPS_CONV1X1=0
# disables, =1 forces all shapes (benching).

import os

import torch.nn as nn
import torch.nn.functional as F

# (cin, cout, stride) -> GEMM path wins fwd+bwd vs MIOpen on MI355X at
# batch 256 (profiles/conv1x1_train_r2b.txt): with MIOpen's wrw kernel
# carrying the weight gradient, the MFMA fwd+dgrad wins 10 of 15
# ResNet-50 1x1 shapes (-10..-17% per conv fwd+bwd).  Still MIOpen:
# the stride-2 downsamples (subsample+scatter overhead) and the
# Cin=64 stem-adjacent shapes.
_GEMM_SHAPES = {
    (128, 512, 1), (256, 64, 1), (256, 128, 1), (256, 1024, 1),
    (512, 128, 1), (512, 256, 1), (512, 2048, 1), (1024, 256, 1),
    (1024, 512, 1), (2048, 512, 1),
}

# (cin, cout) where the hand-written wgrad kernel (csrc/kernels/
# wgrad.hip: dual ds_read_b64_tr_b16 transpose consumption, T14 split
# staging, XCD-clustered block map, split-M fp32 atomics) measured
# faster than MIOpen's wrw (profiles/wgrad_ab_v4.txt); elsewhere
# MIOpen keeps the weight grad.
_WGRAD_SHAPES = {(64, 64), (64, 256), (256, 64), (256, 128), (512, 256)}


def set_gemm_shapes(shapes) -> None:
    _GEMM_SHAPES.clear()
    _GEMM_SHAPES.update(shapes)


def _nt(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """a (M,K) @ b (N,K)^T with the MFMA kernel when the tile contract
    holds, rocBLAS otherwise (GPU), fp32 matmul on CPU."""
    if a.is_cuda and a.dtype == torch.bfloat16:
        m, k = a.shape
        n = b.shape[0]
        if m % 128 == 0 and n % 128 == 0 and k % 64 == 0 and hip_available():
            from rocps.ops import hip_ext
            return hip_ext().gemm_bf16(a, b, None, 0, -1)
        return a @ b.t()
    return (a.float() @ b.float().t()).to(a.dtype)


class _Conv1x1Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, stride):
        n, cin, h, wd = x.shape
        cout = w.shape[0]
        ctx.stride = stride
        ctx.in_shape = x.shape
        xs = x[:, :, ::stride, ::stride] if stride > 1 else x
        xs = xs.contiguous(memory_format=torch.channels_last)
        ho, wo = xs.shape[2], xs.shape[3]
        rows = xs.permute(0, 2, 3, 1).reshape(n * ho * wo, cin)
        wmat = w.reshape(cout, cin)
        ctx.save_for_backward(rows, wmat)
        out = _nt(rows, wmat)
        return (out.reshape(n, ho, wo, cout).permute(0, 3, 1, 2)
                .contiguous(memory_format=torch.channels_last))

    @staticmethod
    def backward(ctx, dy):
        rows, wmat = ctx.saved_tensors
        s = ctx.stride
        n, cin, h, wd = ctx.in_shape
        cout = wmat.shape[0]
        dyc = dy.contiguous(memory_format=torch.channels_last)
        dyr = dyc.permute(0, 2, 3, 1).reshape(-1, cout)
        ho = (h + s - 1) // s
        wo = (wd + s - 1) // s
        # dgrad: dx_rows[m, cin] = sum_cout dy[m, cout] * W[cout, cin]
        dxr = _nt(dyr, wmat.t().contiguous())
        wg = os.environ.get("ROCPS_WGRAD", "auto")
        if (rows.is_cuda and wg != "miopen"
                and (wg == "custom" or (cin, cout) in _WGRAD_SHAPES)
                and rows.dtype == torch.bfloat16):
            # hand-written tall-skinny wgrad (both operands consumed
            # via hardware-transpose reads, split-M fp32 atomics)
            from rocps.ops import hip_ext
            dw = hip_ext().conv1x1_wgrad(dyr, rows).to(wmat.dtype)
        elif rows.is_cuda:
            # MIOpen's wrw kernel (the fwd ran on the stride-1
            # subsampled xs, so wrw of a stride-1 1x1 conv over xs is
            # exactly dW)
            xs = (rows.reshape(n, ho, wo, cin).permute(0, 3, 1, 2)
                  .contiguous(memory_format=torch.channels_last))
            _, dw, _ = torch.ops.aten.convolution_backward(
                dyc, xs, wmat.reshape(cout, cin, 1, 1), None,
                [1, 1], [0, 0], [1, 1], False, [0, 0], 1,
                [False, True, False])
            dw = dw.reshape(cout, cin)
        else:
            dw = (dyr.float().t() @ rows.float()).to(wmat.dtype)
        dxs = (dxr.reshape(n, ho, wo, cin).permute(0, 3, 1, 2)
               .contiguous(memory_format=torch.channels_last))
        if s > 1:
            dx = torch.zeros(ctx.in_shape, dtype=dy.dtype, device=dy.device
                             ).contiguous(memory_format=torch.channels_last)
            dx[:, :, ::s, ::s] = dxs
        else:
            dx = dxs
        return dx, dw.reshape(cout, cin, 1, 1).to(wmat.dtype), None


class Conv1x1(nn.Conv2d):
    """Drop-in nn.Conv2d(k=1, bias=False) that runs the MFMA-GEMM
    train path on shapes where it measured faster than MIOpen
    (tools/bench_conv1x1.py); state_dict-compatible with nn.Conv2d."""

    def __init__(self, cin: int, cout: int, stride: int = 1):
        super().__init__(cin, cout, 1, stride=stride, bias=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        mode = os.environ.get("ROCPS_CONV1X1", "auto")
        if (mode != "0" and x.is_cuda and x.dtype == torch.bfloat16
                and (mode == "1"
                     or (self.in_channels, self.out_channels,
                         self.stride[0]) in _GEMM_SHAPES)):
            return _Conv1x1Fn.apply(x, self.weight, self.stride[0])
        return super().forward(x)
