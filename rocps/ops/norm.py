"""Fused BatchNorm(+residual-add)(+ReLU) — the BN slice of C9.

Reference: TF ``FusedBatchNormV3``/``Grad`` + ``AddV2`` + ``Relu`` on
every conv output of ResNet/W-ResNet. Both directions run as
hand-written CDNA4 kernels on NHWC bf16/fp32 tensors
(csrc/kernels/bn_relu.hip):

- fwd train: channel-group stats kernel + per-channel finalize (one
  C-thread launch; keeps all small math off the host) + one vectorized
  apply ``y = relu(x*scale[c]+shift[c] [+ residual])``.
- bwd: one masked reduction + finalize + one apply producing dx (and
  d_residual = relu-masked dy when the bottleneck's skip-add is fused).

CPU path: plain fp32 PyTorch reference (same math; what the GPU
numerics tests compare against). Training stats are biased variance;
running-var update is unbiased, matching torch BatchNorm2d.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from rocps.ops import hip_ext, require_hip_for

__all__ = ["BatchNormAct2d", "fused_bn_act"]


class _FusedBNAct(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, residual, weight, bias, running_mean, running_var,
                training: bool, momentum: float, eps: float, relu: bool):
        has_res = residual is not None
        ctx.relu = relu
        ctx.has_res = has_res
        ctx.training = training
        if require_hip_for(x):
            assert x.dim() == 4
            xc = x.contiguous(memory_format=torch.channels_last)
            res = (residual.contiguous(memory_format=torch.channels_last)
                   if has_res else None)
            if training:
                y, save_mean, save_invstd, fold4c = \
                    hip_ext().bn_relu_fwd_train(
                        xc, weight, bias, running_mean, running_var,
                        momentum, eps, relu, res)
            else:
                y = hip_ext().bn_relu_fwd_eval(
                    xc, weight, bias, running_mean, running_var, eps, relu,
                    res)
                save_mean = running_mean.float()
                save_invstd = (running_var.float() + eps).rsqrt()
                fold4c = None
            ctx.save_for_backward(xc, weight, save_mean, save_invstd, y,
                                  *( [fold4c] if fold4c is not None else []))
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
                running_mean.mul_(1 - momentum).add_(
                    mean.to(running_mean.dtype), alpha=momentum)
                running_var.mul_(1 - momentum).add_(
                    (var * n / max(n - 1, 1)).to(running_var.dtype),
                    alpha=momentum)
        else:
            mean = running_mean.float()
            var = running_var.float()
        invstd = (var + eps).rsqrt()
        xhat = (x32 - mean[None, :, None, None]) * invstd[None, :, None, None]
        y = xhat * weight.float()[None, :, None, None] \
            + bias.float()[None, :, None, None]
        if has_res:
            y = y + residual.float()
        if relu:
            y = y.relu()
        y = y.to(x.dtype)
        ctx.save_for_backward(x, weight, mean, invstd, y)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, mean, invstd, y = ctx.saved_tensors[:5]
        fold4c = ctx.saved_tensors[5] if len(ctx.saved_tensors) > 5 else None
        if not ctx.training:
            # Eval-mode (frozen-BN) backward: mean/var are CONSTANTS
            # (running stats), so dx has no batch-statistic correction
            # terms — dx = dy * gamma * invstd, relu/residual-masked.
            # Cold path (fine-tuning with frozen BN), computed with
            # plain tensor ops on whatever device x lives on.
            dy32 = dy.float()
            if ctx.relu:
                dy32 = dy32 * (y > 0).float()
            dres = dy32.to(dy.dtype) if ctx.has_res else None
            bc = (None, slice(None), None, None)
            xhat = (x.float() - mean[bc]) * invstd[bc]
            dbeta = dy32.sum(dim=(0, 2, 3))
            dgamma = (dy32 * xhat).sum(dim=(0, 2, 3))
            dx = dy32 * (weight.float() * invstd)[bc]
            return (dx.to(x.dtype), dres, dgamma.to(weight.dtype),
                    dbeta.to(weight.dtype),
                    None, None, None, None, None, None)
        if ctx.hip:
            dyc = dy.contiguous(memory_format=torch.channels_last)
            outs = hip_ext().bn_relu_bwd(
                x, dyc, y, weight, mean, invstd, ctx.relu, ctx.has_res,
                fold4c)
            dx, dgamma, dbeta = outs[0], outs[1], outs[2]
            dres = outs[3] if ctx.has_res else None
            return (dx, dres, dgamma, dbeta,
                    None, None, None, None, None, None)
        dy32 = dy.float()
        if ctx.relu:
            dy32 = dy32 * (y > 0).float()
        dres = dy32.to(dy.dtype) if ctx.has_res else None
        x32 = x.float()
        M = x.numel() / x.shape[1]
        xhat = (x32 - mean[None, :, None, None]) * invstd[None, :, None, None]
        dbeta = dy32.sum(dim=(0, 2, 3))
        dgamma = (dy32 * xhat).sum(dim=(0, 2, 3))
        g = weight.float()[None, :, None, None] * invstd[None, :, None, None]
        dx = g * (dy32 - dbeta[None, :, None, None] / M
                  - xhat * dgamma[None, :, None, None] / M)
        return (dx.to(x.dtype), dres, dgamma.to(weight.dtype),
                dbeta.to(weight.dtype), None, None, None, None, None, None)


def fused_bn_act(x, weight, bias, running_mean, running_var,
                 training: bool, momentum: float = 0.1, eps: float = 1e-5,
                 relu: bool = True,
                 residual: Optional[torch.Tensor] = None) -> torch.Tensor:
    return _FusedBNAct.apply(x, residual, weight, bias, running_mean,
                             running_var, training, momentum, eps, relu)


class BatchNormAct2d(nn.Module):
    """Drop-in BatchNorm2d with fused optional residual-add + ReLU
    epilogue (``act``: "relu", "none", "add_relu").

    State-dict layout matches nn.BatchNorm2d (weight/bias/running_mean/
    running_var/num_batches_tracked) so checkpoints interop.
    """

    def __init__(self, num_features: int, eps: float = 1e-5,
                 momentum: float = 0.1, act: str = "relu"):
        super().__init__()
        assert act in ("relu", "none", "add_relu")
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
        # counted host-side: the buffer lives on GPU with the model and
        # a per-forward `+= 1` is one tiny kernel launch per BN layer
        # per step (53/step on ResNet-50); the python counter folds
        # into the buffer at state_dict/checkpoint time
        self._nbt_pending = 0

    def _save_to_state_dict(self, destination, prefix, keep_vars):
        self._flush_nbt()
        super()._save_to_state_dict(destination, prefix, keep_vars)

    def _flush_nbt(self) -> None:
        if self._nbt_pending:
            with torch.no_grad():
                self.num_batches_tracked += self._nbt_pending
            self._nbt_pending = 0

    def _load_from_state_dict(self, *args, **kw):
        self._nbt_pending = 0
        super()._load_from_state_dict(*args, **kw)

    def forward(self, x: torch.Tensor,
                residual: Optional[torch.Tensor] = None) -> torch.Tensor:
        if self.training:
            self._nbt_pending += 1
        if self.act == "add_relu":
            assert residual is not None, "add_relu needs a residual input"
        return fused_bn_act(x, self.weight, self.bias, self.running_mean,
                            self.running_var, self.training, self.momentum,
                            self.eps, relu=(self.act != "none"),
                            residual=residual)

    def extra_repr(self) -> str:
        return f"{self.num_features}, act={self.act}"
