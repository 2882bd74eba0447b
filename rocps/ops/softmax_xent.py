"""Fused softmax-cross-entropy — capability C11.

Reference: TF ``SoftmaxCrossEntropyWithLogits`` (rowwise online
max/exp/sum + grad in one op). HIP kernel: one workgroup per row,
online softmax, loss + dlogits produced in a single pass
(csrc/kernels/softmax_xent.hip); vocab-split path for large C
(BERT MLM, C=30522). CPU path = plain fp32 reference.
"""

from __future__ import annotations

import torch

from rocps.ops import require_hip_for, hip_ext

__all__ = ["softmax_cross_entropy_with_logits"]


class _SoftmaxXentFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits: torch.Tensor, labels: torch.Tensor):
        if require_hip_for(logits):
            loss, dlogits = hip_ext().softmax_xent_fwd(
                logits.contiguous(), labels.contiguous())
        else:
            l32 = logits.float()
            m = l32.max(dim=-1, keepdim=True).values
            z = (l32 - m).exp()
            s = z.sum(dim=-1, keepdim=True)
            logp = l32 - m - s.log()
            loss = -logp.gather(-1, labels.unsqueeze(-1).long()).squeeze(-1)
            dlogits = (z / s)
            dlogits.scatter_add_(
                -1, labels.unsqueeze(-1).long(),
                -torch.ones_like(loss).unsqueeze(-1))
            dlogits = dlogits.to(logits.dtype)
        ctx.save_for_backward(dlogits)
        return loss

    @staticmethod
    def backward(ctx, grad_loss: torch.Tensor):
        (dlogits,) = ctx.saved_tensors
        return dlogits * grad_loss.unsqueeze(-1).to(dlogits.dtype), None


def softmax_cross_entropy_with_logits(logits: torch.Tensor,
                                      labels: torch.Tensor) -> torch.Tensor:
    """Per-row loss (same contract as the TF op): logits (N, C) any float
    dtype, labels (N,) int64 class indices -> loss (N,) float32."""
    return _SoftmaxXentFn.apply(logits, labels)
