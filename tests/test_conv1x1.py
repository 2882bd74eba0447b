"""conv1x1-as-GEMM plumbing (ROADMAP §Perf 1 staging): shape, stride
and tile-padding logic verified against torch's conv2d on CPU; the GPU
bf16 path reuses the already-numerics-tested MFMA GEMM kernel."""

import pytest
import torch
import torch.nn.functional as F

from rocps.ops.conv1x1 import conv1x1_nhwc, forward_supported


@pytest.mark.parametrize("shape,stride", [
    ((2, 64, 14, 14), 1),    # bottleneck conv1-like
    ((2, 256, 14, 14), 1),   # bottleneck conv3-like
    ((2, 256, 14, 14), 2),   # downsample path
    ((1, 512, 7, 7), 1),
    ((3, 64, 9, 9), 2),      # odd spatial + stride
])
def test_conv1x1_matches_torch(shape, stride):
    g = torch.Generator().manual_seed(0)
    n, c, h, w = shape
    cout = 96  # deliberately NOT a tile multiple: exercises N-padding
    x = torch.randn(n, c, h, w, generator=g).contiguous(
        memory_format=torch.channels_last)
    wt = torch.randn(cout, c, 1, 1, generator=g)
    ref = F.conv2d(x, wt, stride=stride)
    got = conv1x1_nhwc(x, wt, stride=stride)
    assert got.shape == ref.shape
    assert torch.allclose(got, ref, atol=1e-4, rtol=1e-4)


def test_conv1x1_rejects_unsupported_cin():
    assert not forward_supported(48)
    x = torch.randn(1, 48, 8, 8)
    with pytest.raises(ValueError):
        conv1x1_nhwc(x, torch.randn(64, 48, 1, 1))


def test_conv1x1_padding_math():
    """M and N padding round up to the 128 tile and slice back."""
    from rocps.ops.conv1x1 import _pad_to
    t = torch.ones(100, 64)
    p = _pad_to(t, 0, 128)
    assert p.shape == (128, 64)
    assert p[100:].abs().sum() == 0
    assert _pad_to(p, 0, 128) is p  # already aligned: no copy


@pytest.mark.parametrize("shape,stride", [
    ((2, 64, 8, 8), 1),
    ((2, 128, 8, 8), 2),
])
def test_conv1x1_module_autograd_matches_conv2d(shape, stride):
    """The Conv1x1 train path (GEMM fwd + GEMM dgrad + TN wgrad) against
    autograd through F.conv2d, forward AND gradients, incl. the stride-2
    scatter in dgrad."""
    import os

    from rocps.ops.conv1x1 import _Conv1x1Fn

    g = torch.Generator().manual_seed(1)
    n, c, h, wd = shape
    cout = 64
    x = torch.randn(n, c, h, wd, generator=g).contiguous(
        memory_format=torch.channels_last).requires_grad_(True)
    w = torch.randn(cout, c, 1, 1, generator=g).requires_grad_(True)
    y = _Conv1x1Fn.apply(x, w, stride)
    gy = torch.randn(y.shape, generator=g)
    y.backward(gy)

    x2 = x.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    ref = F.conv2d(x2, w2, stride=stride)
    ref.backward(gy)
    assert torch.allclose(y, ref, atol=1e-4, rtol=1e-4)
    assert torch.allclose(x.grad, x2.grad, atol=1e-4, rtol=1e-4)
    assert torch.allclose(w.grad, w2.grad, atol=1e-3, rtol=1e-3)


def test_conv1x1_module_state_dict_compat():
    from rocps.ops.conv1x1 import Conv1x1
    m = Conv1x1(64, 128, stride=2)
    ref = torch.nn.Conv2d(64, 128, 1, stride=2, bias=False)
    ref.load_state_dict(m.state_dict())   # same keys/shapes
    x = torch.randn(2, 64, 8, 8)
    assert torch.allclose(m(x), ref(x), atol=1e-5)
