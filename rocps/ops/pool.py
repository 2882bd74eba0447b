"""MaxPool 3x3/stride-2/pad-1 — the ResNet stem pool (C9's pooling
slice; reference: TF MaxPool/MaxPoolGrad). HIP path stores 1-byte
window codes instead of torch's int64 indices and the backward is a
deterministic gather (csrc/kernels/maxpool.hip). CPU path: torch."""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from rocps.ops import hip_ext, require_hip_for

__all__ = ["MaxPool332"]


class _MaxPool332Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        xc = x.contiguous(memory_format=torch.channels_last)
        y, idx = hip_ext().maxpool332_fwd(xc)
        ctx.save_for_backward(idx)
        ctx.in_hw = (x.shape[2], x.shape[3])
        return y

    @staticmethod
    def backward(ctx, dy):
        (idx,) = ctx.saved_tensors
        H, W = ctx.in_hw
        return hip_ext().maxpool332_bwd(dy, idx, H, W)


class MaxPool332(nn.Module):
    """Drop-in for nn.MaxPool2d(3, stride=2, padding=1) on NHWC GPUs."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if x.is_cuda and x.shape[1] % 8 == 0:
            require_hip_for(x)
            return _MaxPool332Fn.apply(x)
        return F.max_pool2d(x, 3, stride=2, padding=1)
