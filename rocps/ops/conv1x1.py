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
plain fp32 matmul, which is what the parity tests pin down. Dispatch
into the model forward stays OFF until the GPU measurement pass
(round 2): MIOpen's igemm asm is the measured baseline to beat
per-shape [profiles/RESULTS.md].
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
