"""Fused LayerNorm — BERT hot op (SURVEY §2d `bert_ops`: fused LN
fwd/bwd on (B·S, 768) rows). HIP path: csrc/kernels/layernorm.hip;
CPU path: fp32 reference."""

from __future__ import annotations

import torch
import torch.nn as nn

from rocps.ops import hip_ext, require_hip_for

__all__ = ["LayerNorm", "layer_norm"]


class _LayerNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        shape = x.shape
        x2 = x.reshape(-1, shape[-1])
        if require_hip_for(x2):
            w32 = weight.detach().float().contiguous()
            b32 = bias.detach().float().contiguous()
            y, mean, invstd = hip_ext().ln_fwd(x2.contiguous(), w32, b32, eps)
            ctx.save_for_backward(x2, weight, mean, invstd)
            ctx.hip = True
            return y.reshape(shape)
        ctx.hip = False
        x32 = x2.float()
        mean = x32.mean(-1)
        var = x32.var(-1, unbiased=False)
        invstd = (var + eps).rsqrt()
        xhat = (x32 - mean[:, None]) * invstd[:, None]
        y = xhat * weight.float() + bias.float()
        ctx.save_for_backward(x2, weight, mean, invstd)
        return y.to(x.dtype).reshape(shape)

    @staticmethod
    def backward(ctx, dy):
        x2, weight, mean, invstd = ctx.saved_tensors
        D = x2.shape[-1]
        dy2 = dy.reshape(-1, D)
        if ctx.hip:
            w32 = weight.detach().float().contiguous()
            dx, dgamma, dbeta = hip_ext().ln_bwd(x2, dy2.contiguous(), w32,
                                                 mean, invstd)
            return (dx.reshape(dy.shape), dgamma.to(weight.dtype),
                    dbeta.to(weight.dtype), None)
        x32 = x2.float()
        dy32 = dy2.float()
        xhat = (x32 - mean[:, None]) * invstd[:, None]
        g = dy32 * weight.float()
        c1 = (g * xhat).mean(-1)
        c2 = g.mean(-1)
        dx = invstd[:, None] * (g - c2[:, None] - xhat * c1[:, None])
        dgamma = (dy32 * xhat).sum(0)
        dbeta = dy32.sum(0)
        return (dx.to(dy.dtype).reshape(dy.shape), dgamma.to(weight.dtype),
                dbeta.to(weight.dtype), None)


def layer_norm(x, weight, bias, eps: float = 1e-12):
    return _LayerNormFn.apply(x, weight, bias, eps)


class LayerNorm(nn.Module):
    def __init__(self, dim: int, eps: float = 1e-12):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(dim))
        self.bias = nn.Parameter(torch.zeros(dim))
        self.eps = eps

    def forward(self, x):
        return layer_norm(x, self.weight, self.bias, self.eps)
