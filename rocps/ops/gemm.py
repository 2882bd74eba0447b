"""Hand-written MFMA bf16 GEMM — capability C10.

``matmul_nt(a, w)`` computes a @ w.T (the torch Linear contract) on the
CDNA4 MFMA kernels, with optional fused bias and tanh-GELU epilogues —
the SURVEY C10 "bias+GELU epilogue variant for BERT".  Two kernels sit
behind one dispatch (csrc/kernels/gemm_bf16.hip / gemm_bf16_256.hip):

- 128x128 macro-tile, global_load_lds staging, XOR-swizzled LDS,
  16x16x32 bf16 MFMA — 2-3 blocks/CU, wins small grids, shallow K and
  huge grids;
- 256x256 8-phase pipeline (8 waves, one half-tile staged per phase,
  counted s_waitcnt vmcnt across raw barriers, s_setprio'd 16-MFMA
  clusters) — 1 block/CU, wins mid grids at K>=1024 (1.12 PF/s at
  4096^3 vs 0.86 for the 128-tile kernel, profiles/gemm_micro_r2c.txt).

They run on the training hot path: conv1x1 forward+dgrad in
ResNet/W-ResNet (rocps/ops/conv1x1.py) and the BERT MLP fc1 (epilogue
3 emits the pre-activation for backward reuse).  Plain library GEMMs
elsewhere go through rocBLAS via torch.nn.Linear.  Shape contract
M%128 == N%128 == 0, K%64 == 0 (``supported(...)``).
"""

from __future__ import annotations

from typing import Optional

import torch

from rocps.ops import hip_ext

__all__ = ["matmul_nt", "linear_gelu", "linear_gelu_train", "supported"]


def supported(m: int, n: int, k: int) -> bool:
    return m % 128 == 0 and n % 128 == 0 and k % 64 == 0


def matmul_nt(a: torch.Tensor, w: torch.Tensor,
              bias: Optional[torch.Tensor] = None,
              gelu: bool = False) -> torch.Tensor:
    """a (M,K) bf16 @ w (N,K) bf16 -> (M,N) bf16, fp32 accumulation."""
    epi = 2 if gelu else (1 if bias is not None else 0)
    return hip_ext().gemm_bf16(a, w, bias, epi)


class _LinearGeluFn(torch.autograd.Function):
    """Fused fwd (MFMA GEMM + bias + GELU); backward recomputes the
    pre-activation with a library GEMM (plain GEMMs may use rocBLAS)."""

    @staticmethod
    def forward(ctx, x, w, b):
        ctx.save_for_backward(x, w, b)
        return matmul_nt(x, w, b, gelu=True)

    @staticmethod
    def backward(ctx, dy):
        x, w, b = ctx.saved_tensors
        pre = (x @ w.t()).float() + b.float()
        c = 0.7978845608028654
        t = torch.tanh(c * (pre + 0.044715 * pre ** 3))
        dgelu = 0.5 * (1 + t) + 0.5 * pre * (1 - t * t) * c * (
            1 + 3 * 0.044715 * pre ** 2)
        g = (dy.float() * dgelu).to(dy.dtype)
        dx = g @ w
        dw = g.t() @ x
        db = g.sum(0)
        return dx, dw, db.to(b.dtype)


def linear_gelu(x: torch.Tensor, w: torch.Tensor,
                b: torch.Tensor) -> torch.Tensor:
    return _LinearGeluFn.apply(x, w, b)


def _dgelu_tanh(pre: torch.Tensor) -> torch.Tensor:
    c = 0.7978845608028654
    t = torch.tanh(c * (pre + 0.044715 * pre ** 3))
    return 0.5 * (1 + t) + 0.5 * pre * (1 - t * t) * c * (
        1 + 3 * 0.044715 * pre ** 2)


class _LinearGeluTrainFn(torch.autograd.Function):
    """Training variant (ROADMAP §Perf 5): the fused kernel emits the
    pre-activation as a second output (epilogue 3), so backward reuses
    it instead of the fp32 recompute chain that cost 9 ms/step on
    BERT-base in round 1. Unwired from the model until the round-2 GPU
    numerics + perf pass."""

    @staticmethod
    def forward(ctx, x, w, b):
        if (x.is_cuda and x.dtype == torch.bfloat16
                and supported(x.shape[0], w.shape[0], x.shape[1])):
            from rocps.ops import hip_ext
            y, pre = hip_ext().gemm_bf16_gelu_pre(x, w, b)
        else:  # CPU / reference path: same math, library ops
            pre = (x.float() @ w.float().t() + b.float()).to(x.dtype)
            p32 = pre.float()
            c = 0.7978845608028654
            y = (0.5 * p32 * (1 + torch.tanh(
                c * (p32 + 0.044715 * p32 ** 3)))).to(x.dtype)
        ctx.save_for_backward(x, w, b, pre)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, b, pre = ctx.saved_tensors
        if dy.is_cuda and dy.dtype == torch.bfloat16:
            # one-pass fused dy * dgelu(pre) (csrc/kernels/gelu.hip) —
            # the composed torch chain materializes several fp32
            # (B*S, inter) intermediates per layer
            from rocps.ops import hip_ext
            g = hip_ext().gelu_bwd(pre, dy.contiguous())
        else:
            g = (dy.float() * _dgelu_tanh(pre.float())).to(dy.dtype)
        dx = g @ w
        dw = g.t() @ x
        db = g.float().sum(0)
        return dx, dw, db.to(b.dtype)


def linear_gelu_train(x: torch.Tensor, w: torch.Tensor,
                      b: torch.Tensor) -> torch.Tensor:
    return _LinearGeluTrainFn.apply(x, w, b)
