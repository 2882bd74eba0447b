"""Fused optimizer updates — capability C8.

Reference: TF's per-variable ``ApplyGradientDescent`` /
``ApplyMomentum`` / ``ApplyAdam`` kernels, one launch per variable on
the PS device. Here: one grid-stride HIP kernel per *flat bucket/shard*
(csrc/kernels/fused_optimizers.hip), bf16 params + fp32 master weights
updated in the same pass (single read of grad, single write of both).
"""

from __future__ import annotations

from typing import Optional

import torch

from rocps.ops import hip_available, hip_ext

__all__ = ["available", "sgd", "momentum", "adam"]


def available() -> bool:
    return hip_available()


def sgd(params: torch.Tensor, grads: torch.Tensor,
        master: Optional[torch.Tensor], lr: float) -> None:
    hip_ext().fused_sgd(params, grads,
                        master if master is not None else params, lr,
                        master is not None)


def momentum(params: torch.Tensor, grads: torch.Tensor,
             master: Optional[torch.Tensor], mom_buf: torch.Tensor,
             lr: float, mu: float, nesterov: bool) -> None:
    hip_ext().fused_momentum(params, grads,
                             master if master is not None else params,
                             mom_buf, lr, mu, nesterov, master is not None)


def adam(params: torch.Tensor, grads: torch.Tensor,
         master: Optional[torch.Tensor], m: torch.Tensor, v: torch.Tensor,
         alpha: float, beta1: float, beta2: float, eps: float) -> None:
    hip_ext().fused_adam(params, grads,
                         master if master is not None else params,
                         m, v, alpha, beta1, beta2, eps, master is not None)
