"""Model zoo construction + forward/loss shape checks (SURVEY §2a:
the reference's model set rebuilt — MNIST MLP, ResNet-50,
Wide-ResNet-101-2, BERT-base). Heavy models run one tiny CPU forward;
full training steps are covered by the engine/GPU tests.
"""

import pytest
import torch

from rocps.models import build_model


def _n_params(m):
    return sum(p.numel() for p in m.parameters())


def test_build_model_unknown():
    with pytest.raises(ValueError):
        build_model("alexnet")


def test_mnist_mlp_loss():
    m = build_model("mnist_mlp", hidden=32)
    x, y = torch.rand(4, 784), torch.randint(0, 10, (4,))
    loss = m.loss(x, y)
    assert loss.ndim == 0 and torch.isfinite(loss)


def test_resnet50_shape():
    m = build_model("resnet50")
    # torchvision-style resnet50: 25.56 M params, 1000-way head
    assert abs(_n_params(m) - 25_557_032) < 20_000
    m.eval()
    with torch.no_grad():
        out = m(torch.randn(1, 3, 64, 64))
    assert out.shape == (1, 1000)
    loss = m.loss(torch.randn(2, 3, 64, 64), torch.randint(0, 1000, (2,)))
    assert torch.isfinite(loss)


def test_wide_resnet101_2_shape():
    m = build_model("wide_resnet101_2")
    # 126.89 M params (torchvision wide_resnet101_2)
    assert abs(_n_params(m) - 126_886_696) < 50_000
    m.eval()
    with torch.no_grad():
        out = m(torch.randn(1, 3, 64, 64))
    assert out.shape == (1, 1000)


def test_bert_base_shape():
    m = build_model("bert_base")
    # BERT-base: ~110 M params incl. MLM head tie
    n = _n_params(m)
    assert 100e6 < n < 135e6, n
    ids = torch.randint(0, 30522, (2, 16))
    lbl = torch.randint(0, 30522, (2, 16))
    loss = m.loss(ids, lbl)
    assert torch.isfinite(loss)


def test_bn_eval_mode_backward_matches_frozen_stats():
    """eval-mode (frozen-BN) backward must treat
    running mean/var as CONSTANTS — dx = dy * gamma * invstd with no
    batch-statistic correction terms. Compare against autograd through
    F.batch_norm(training=False)."""
    import torch.nn.functional as F

    from rocps.ops.norm import fused_bn_act

    torch.manual_seed(0)
    N, C, H, W = 4, 8, 5, 5
    for relu in (False, True):
        x = torch.randn(N, C, H, W, dtype=torch.float32, requires_grad=True)
        w = torch.randn(C, requires_grad=True)
        b = torch.randn(C, requires_grad=True)
        rm = torch.randn(C)
        rv = torch.rand(C) + 0.5
        y = fused_bn_act(x, w, b, rm.clone(), rv.clone(),
                         training=False, relu=relu)
        gy = torch.randn_like(y)
        y.backward(gy)

        x2 = x.detach().clone().requires_grad_(True)
        w2 = w.detach().clone().requires_grad_(True)
        b2 = b.detach().clone().requires_grad_(True)
        ref = F.batch_norm(x2, rm.clone(), rv.clone(), w2, b2,
                           training=False, eps=1e-5)
        if relu:
            ref = ref.relu()
        ref.backward(gy)

        assert torch.allclose(y, ref, atol=1e-5), (relu,)
        assert torch.allclose(x.grad, x2.grad, atol=1e-5), (relu,)
        assert torch.allclose(w.grad, w2.grad, atol=1e-4), (relu,)
        assert torch.allclose(b.grad, b2.grad, atol=1e-4), (relu,)


def test_bn_num_batches_tracked_lazy_counter():
    """num_batches_tracked counts host-side per forward and folds into
    the buffer at state_dict time (no per-layer GPU launch per step)."""
    from rocps.ops.norm import BatchNormAct2d
    m = BatchNormAct2d(4, act="none")
    x = torch.randn(2, 4, 3, 3)
    for _ in range(5):
        m(x)
    assert m._nbt_pending == 5
    sd = m.state_dict()
    assert int(sd["num_batches_tracked"]) == 5
    assert m._nbt_pending == 0
    m(x)
    sd = m.state_dict()
    assert int(sd["num_batches_tracked"]) == 6
    # load resets the pending counter
    m2 = BatchNormAct2d(4, act="none")
    m2.load_state_dict(sd)
    assert int(m2.num_batches_tracked) == 6 and m2._nbt_pending == 0
