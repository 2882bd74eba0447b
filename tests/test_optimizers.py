"""Flat-apply optimizers vs torch.optim references (capability C8)."""

import pytest
import torch

from rocps.train.optimizer import (AdamOptimizer, GradientDescentOptimizer,
                                   MomentumOptimizer)


def _run_flat(opt, p0, grads_seq):
    p = p0.clone()
    for g in grads_seq:
        if hasattr(opt, "begin_step"):
            opt.begin_step()
        opt.apply_flat(p, g, slot="s")
    return p


def _grads(n, steps, seed=0):
    g = torch.Generator().manual_seed(seed)
    return [torch.randn(n, generator=g) for _ in range(steps)]


def test_sgd_matches_torch():
    p0 = torch.randn(257)
    gs = _grads(257, 5)
    ours = _run_flat(GradientDescentOptimizer(0.1), p0, gs)
    ref = p0.clone().requires_grad_(True)
    topt = torch.optim.SGD([ref], lr=0.1)
    for g in gs:
        ref.grad = g.clone()
        topt.step()
    assert torch.allclose(ours, ref.detach(), atol=1e-6)


def test_momentum_matches_torch():
    # TF ApplyMomentum: v = mu*v + g; p -= lr*v  == torch SGD(momentum=mu,
    # dampening=0, weight_decay=0)
    p0 = torch.randn(100)
    gs = _grads(100, 8, seed=1)
    ours = _run_flat(MomentumOptimizer(0.05, 0.9), p0, gs)
    ref = p0.clone().requires_grad_(True)
    topt = torch.optim.SGD([ref], lr=0.05, momentum=0.9)
    for g in gs:
        ref.grad = g.clone()
        topt.step()
    assert torch.allclose(ours, ref.detach(), atol=1e-6)


def test_nesterov_momentum_matches_torch():
    p0 = torch.randn(64)
    gs = _grads(64, 6, seed=2)
    ours = _run_flat(MomentumOptimizer(0.05, 0.9, use_nesterov=True), p0, gs)
    ref = p0.clone().requires_grad_(True)
    topt = torch.optim.SGD([ref], lr=0.05, momentum=0.9, nesterov=True)
    for g in gs:
        ref.grad = g.clone()
        topt.step()
    assert torch.allclose(ours, ref.detach(), atol=1e-6)


def test_adam_tf_semantics():
    """TF Adam: alpha_t = lr*sqrt(1-b2^t)/(1-b1^t); p -= alpha_t*m/(sqrt(v)+eps).

    This equals torch.optim.Adam's update except torch divides by
    (sqrt(v_hat)+eps) with v_hat=v/(1-b2^t) — i.e. eps placement differs
    by the sqrt(1-b2^t) factor. Verify against an explicit reference.
    """
    lr, b1, b2, eps = 1e-3, 0.9, 0.999, 1e-8
    p0 = torch.randn(50)
    gs = _grads(50, 10, seed=3)
    ours = _run_flat(AdamOptimizer(lr, b1, b2, eps), p0, gs)
    p = p0.clone()
    m = torch.zeros(50)
    v = torch.zeros(50)
    for t, g in enumerate(gs, start=1):
        m = b1 * m + (1 - b1) * g
        v = b2 * v + (1 - b2) * g * g
        alpha = lr * (1 - b2 ** t) ** 0.5 / (1 - b1 ** t)
        p = p - alpha * m / (v.sqrt() + eps)
    assert torch.allclose(ours, p, atol=1e-6)


def test_bf16_master_weights():
    """bf16 params keep an fp32 master: many tiny updates must not be lost
    to bf16 rounding (the torch-only bf16 path would stall)."""
    p = (torch.ones(32)).bfloat16()
    opt = GradientDescentOptimizer(1e-3)
    g = torch.full((32,), 0.01, dtype=torch.bfloat16)
    for _ in range(100):
        opt.apply_flat(p, g, slot="s")
    # 100 * 1e-3 * 0.01 = 1e-3 total decrement, far below bf16 ulp at 1.0
    master = opt._state["s"]["master"]
    assert torch.allclose(master, torch.full((32,), 1 - 1e-3), atol=1e-5)
    # bf16 view tracks master's rounded value
    assert torch.equal(p, master.bfloat16())


def test_state_per_slot_isolated():
    opt = MomentumOptimizer(0.1, 0.9)
    a = torch.zeros(4)
    b = torch.zeros(4)
    opt.apply_flat(a, torch.ones(4), slot="A")
    opt.apply_flat(b, torch.full((4,), 2.0), slot="B")
    assert not torch.equal(opt._state["A"]["mom"], opt._state["B"]["mom"])


def test_cpu_fused_apply_matches_torch_path(monkeypatch):
    """csrc/cpu/cpu_apply.cc (async PS apply, ROADMAP §Perf 4): the
    fused single-pass loops match the torch-op path — bitwise for
    SGD/momentum (FMA rounding mirrored), ~1 ulp for Adam (torch's
    SLEEF sqrt is not correctly rounded)."""
    pytest.importorskip("rocps._cpuapply")
    import rocps.train.optimizer as O

    n = 4097  # odd size: exercises any vector tail
    for make, exact in [
        (lambda: O.GradientDescentOptimizer(0.17), True),
        (lambda: O.MomentumOptimizer(0.1, 0.9), True),
        (lambda: O.MomentumOptimizer(0.1, 0.9, use_nesterov=True), True),
        (lambda: O.AdamOptimizer(0.01), False),
    ]:
        g = torch.Generator().manual_seed(0)
        grads = torch.randn(n, generator=g)
        results = {}
        for mode in ("fused", "torch"):
            monkeypatch.setenv("ROCPS_CPU_FUSED",
                               "1" if mode == "fused" else "0")
            opt = make()
            if hasattr(opt, "_t"):
                opt._t = 3
            p = torch.linspace(-2, 2, n)
            for _ in range(5):
                opt.apply_flat(p, grads, slot=0)
            results[mode] = p
        if exact:
            assert torch.equal(results["fused"], results["torch"]), make
        else:
            assert torch.allclose(results["fused"], results["torch"],
                                  atol=1e-5, rtol=1e-5)


def test_cpu_fused_apply_bf16_master(monkeypatch):
    """bf16 shard + fp32 master through the fused CPU path equals the
    torch-op master-weight path."""
    pytest.importorskip("rocps._cpuapply")
    import rocps.train.optimizer as O

    n = 1031
    g = torch.Generator().manual_seed(1)
    grads = torch.randn(n, generator=g).bfloat16()
    results = {}
    for mode in ("fused", "torch"):
        monkeypatch.setenv("ROCPS_CPU_FUSED",
                           "1" if mode == "fused" else "0")
        opt = O.MomentumOptimizer(0.2, 0.9)
        p = torch.linspace(-1, 1, n).bfloat16()
        for _ in range(6):
            opt.apply_flat(p, grads, slot=0)
        results[mode] = (p, opt._state[0]["master"].clone())
    assert torch.equal(results["fused"][0], results["torch"][0])
    assert torch.equal(results["fused"][1], results["torch"][1])
