"""CPU tests for the fused-LayerNorm op (reference path) and BERT."""

import torch

from rocps.models.bert import bert_base
from rocps.ops.layer_norm import layer_norm


def test_layernorm_matches_torch():
    torch.manual_seed(0)
    x = torch.randn(10, 320, requires_grad=True)
    w = (torch.rand(320) + 0.5).requires_grad_(True)
    b = torch.randn(320).requires_grad_(True)
    y = layer_norm(x, w, b, 1e-12)
    ref = torch.nn.functional.layer_norm(x, (320,), w, b, 1e-12)
    assert torch.allclose(y, ref, atol=1e-5)
    g = torch.randn_like(y)
    y.backward(g, retain_graph=False)
    gx, gw, gb = x.grad.clone(), w.grad.clone(), b.grad.clone()
    x.grad = w.grad = b.grad = None
    ref2 = torch.nn.functional.layer_norm(x, (320,), w, b, 1e-12)
    ref2.backward(g)
    assert torch.allclose(gx, x.grad, atol=1e-5)
    assert torch.allclose(gw, w.grad, atol=1e-4)
    assert torch.allclose(gb, b.grad, atol=1e-4)


def test_layernorm_3d_shape():
    x = torch.randn(2, 5, 64)
    w = torch.ones(64)
    b = torch.zeros(64)
    y = layer_norm(x, w, b)
    assert y.shape == x.shape
    assert torch.allclose(y.mean(-1), torch.zeros(2, 5), atol=1e-5)


def test_bert_tiny_train_step():
    from rocps.engine.comm import Comm
    from rocps.engine.sync_engine import SyncEngine
    from rocps.train.optimizer import AdamOptimizer
    torch.manual_seed(1)
    m = bert_base(layers=2, hidden=64, heads=4, intermediate=128, vocab=300,
                  max_pos=32)
    eng = SyncEngine(m, Comm(), bucket_cap_mb=1.0)
    opt = AdamOptimizer(1e-3)
    ids = torch.randint(0, 300, (4, 16))
    lbl = torch.randint(0, 300, (4, 16))
    losses = []
    for _ in range(8):
        eng.zero_grad()
        opt.begin_step()
        loss = m.loss(ids, lbl)
        eng.step(loss, opt)
        losses.append(float(loss.detach()))
    assert losses[-1] < losses[0]  # overfits the fixed batch


def test_linear_gelu_train_grads_match_autograd():
    """linear_gelu_train (fused epilogue-3 path staged for round 2):
    the saved-pre backward matches plain autograd on the CPU reference
    path (same math the GPU kernel implements)."""
    import torch

    from rocps.ops.gemm import linear_gelu_train

    g = torch.Generator().manual_seed(3)
    x = torch.randn(16, 32, generator=g, requires_grad=True)
    w = torch.randn(24, 32, generator=g, requires_grad=True)
    b = torch.randn(24, generator=g, requires_grad=True)

    y = linear_gelu_train(x, w, b)
    y.backward(torch.ones_like(y))
    gx, gw, gb = x.grad.clone(), w.grad.clone(), b.grad.clone()

    x2 = x.detach().clone().requires_grad_()
    w2 = w.detach().clone().requires_grad_()
    b2 = b.detach().clone().requires_grad_()
    pre = x2 @ w2.t() + b2
    c = 0.7978845608028654
    ref = 0.5 * pre * (1 + torch.tanh(c * (pre + 0.044715 * pre ** 3)))
    assert torch.allclose(y, ref, atol=1e-5)
    ref.backward(torch.ones_like(ref))
    assert torch.allclose(gx, x2.grad, atol=1e-4)
    assert torch.allclose(gw, w2.grad, atol=1e-4)
    assert torch.allclose(gb, b2.grad, atol=1e-4)
