"""Fused BatchNorm(+ReLU) — the BN part of capability C9.

Reference: TF ``FusedBatchNormV3``/``Grad`` + separate ``Relu`` — on
every conv output of ResNet/W-ResNet. Here both directions run as
hand-written CDNA4 kernels on NHWC bf16/fp32 tensors
(csrc/kernels/bn_relu.hip):

- fwd: one channel-parallel stats kernel (sum/sumsq, fp32 accum,
  per-block partials + device atomics) + one vectorized
  normalize+affine+ReLU kernel → 2 HBM passes instead of 4.
- bwd: one reduction kernel (sum_dy, sum_dy*xhat with the ReLU mask
  folded in) + one apply kernel producing dx → 2 passes instead of 5.

CPU path: plain fp32 PyTorch reference (same math, used by numerics
tests). Training-mode stats are biased variance (TF/torch semantics);
running stats update uses unbiased variance like torch BatchNorm2d.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from rocps.ops import hip_ext, require_hip_for

__all__ = ["BatchNormAct2d", "fused_bn_act"]


class _FusedBNAct(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, running_mean, running_var,
                training: bool, momentum: float, eps: float, relu: bool):
        if require_hip_for(x):
            assert x.dim() == 4
            xc = x.contiguous(memory_format=torch.channels_last)
            if training:
                y, save_mean, save_invstd = hip_ext().bn_relu_fwd_train(
                    xc, weight, bias, running_mean, running_var,
                    momentum, eps, relu)
            else:
                y = hip_ext().bn_relu_fwd_eval(
                    xc, weight, bias, running_mean, running_var, eps, relu)
                save_mean = running_mean.float()
                save_invstd = (running_var.float() + eps).rsqrt()
            ctx.save_for_backward(xc, weight, save_mean, save_invstd, y)
            ctx.relu = relu
            ctx.hip = True
            return y
        # CPU fp32 reference path
        ctx.hip = False
        x32 = x.float()
        if training:
            mean = x32.mean(dim=(0, 2, 3))
            var = x32.var(dim=(0, 2, 3), unbiased=False)
            n = x.numel() / x.shape[1]
            with torch.no_grad():
                running_mean.mul_(1 - momentum).add_(mean, alpha=momentum)
                running_var.mul_(1 - momentum).add_(
                    var * n / max(n - 1, 1), alpha=momentum)
        else:
            mean = running_mean.float()
            var = running_var.float()
        invstd = (var + eps).rsqrt()
        xhat = (x32 - mean[None, :, None, None]) * invstd[None, :, None, None]
        y = xhat * weight.float()[None, :, None, None] \
            + bias.float()[None, :, None, None]
        if relu:
            y = y.relu()
        y = y.to(x.dtype)
        ctx.save_for_backward(x, weight, mean, invstd, y)
        ctx.relu = relu
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, mean, invstd, y = ctx.saved_tensors
        if ctx.hip:
            dyc = dy.contiguous(memory_format=torch.channels_last)
            dx, dgamma, dbeta = hip_ext().bn_relu_bwd(
                x, dyc, y, weight, mean, invstd, ctx.relu)
            return (dx, dgamma, dbeta, None, None, None, None, None, None)
        dy32 = dy.float()
        if ctx.relu:
            dy32 = dy32 * (y > 0).float()
        x32 = x.float()
        M = x.numel() / x.shape[1]
        xhat = (x32 - mean[None, :, None, None]) * invstd[None, :, None, None]
        dbeta = dy32.sum(dim=(0, 2, 3))
        dgamma = (dy32 * xhat).sum(dim=(0, 2, 3))
        g = weight.float()[None, :, None, None] * invstd[None, :, None, None]
        dx = g * (dy32 - dbeta[None, :, None, None] / M
                  - xhat * dgamma[None, :, None, None] / M)
        return (dx.to(x.dtype), dgamma.to(weight.dtype),
                dbeta.to(weight.dtype), None, None, None, None, None, None)


def fused_bn_act(x, weight, bias, running_mean, running_var,
                 training: bool, momentum: float = 0.1, eps: float = 1e-5,
                 relu: bool = True) -> torch.Tensor:
    return _FusedBNAct.apply(x, weight, bias, running_mean, running_var,
                             training, momentum, eps, relu)


class BatchNormAct2d(nn.Module):
    """Drop-in BatchNorm2d with fused optional ReLU epilogue.

    State-dict layout matches nn.BatchNorm2d (weight/bias/running_mean/
    running_var/num_batches_tracked) so checkpoints interop.
    """

    def __init__(self, num_features: int, eps: float = 1e-5,
                 momentum: float = 0.1, act: str = "relu"):
        super().__init__()
        assert act in ("relu", "none")
        self.num_features = num_features
        self.eps = eps
        self.momentum = momentum
        self.act = act
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))
        self.register_buffer("num_batches_tracked",
                             torch.tensor(0, dtype=torch.long))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self.training:
            self.num_batches_tracked += 1
        return fused_bn_act(x, self.weight, self.bias, self.running_mean,
                            self.running_var, self.training, self.momentum,
                            self.eps, relu=(self.act == "relu"))

    def extra_repr(self) -> str:
        return f"{self.num_features}, act={self.act}"
