"""Sync-engine unit tests (single process, world=1) — C4/C7 logic."""

import torch

from rocps.engine.buckets import build_buckets
from rocps.engine.comm import Comm
from rocps.engine.sync_engine import SyncEngine
from rocps.models import mnist_mlp, synthetic_mnist_batch
from rocps.train.optimizer import GradientDescentOptimizer, MomentumOptimizer


def test_buckets_structure():
    m = mnist_mlp(hidden=64)
    buckets = build_buckets(m.named_parameters(), world_size=2,
                            bucket_cap_mb=0.05, flatten_params=True)
    assert len(buckets) >= 2  # cap forces multiple buckets
    total = sum(p.numel() for p in m.parameters())
    assert sum(b.numel for b in buckets) == total
    for b in buckets:
        assert b.numel_padded % (2 * 64) == 0
        assert b.shard_numel * 2 == b.numel_padded
        # params were re-pointed into flat storage
        for (name, p), off in zip(b.named_params, b.offsets):
            assert p.data.data_ptr() == b.params_flat[off:].data_ptr()
            assert p.grad.data_ptr() == b.grads[off:].data_ptr()


def test_buckets_fuzz_invariants():
    """Property fuzz: arbitrary parameter shapes / world sizes keep the
    bucket invariants — full coverage, world*64 padding, shard split,
    view aliasing, value preservation through the repack."""
    from hypothesis import given, settings, strategies as st

    @settings(deadline=None, max_examples=25)
    @given(
        shapes=st.lists(st.lists(st.integers(1, 9), min_size=1,
                                 max_size=3), min_size=1, max_size=10),
        world=st.sampled_from([1, 2, 4, 8]),
        cap_kb=st.sampled_from([1, 4, 64]),
    )
    def check(shapes, world, cap_kb):
        params = [(f"p{i}", torch.nn.Parameter(torch.randn(s)))
                  for i, s in enumerate(shapes)]
        originals = {n: p.detach().clone() for n, p in params}
        buckets = build_buckets(params, world_size=world,
                                bucket_cap_mb=cap_kb / 1024,
                                flatten_params=True)
        total = sum(v.numel() for v in originals.values())
        assert sum(b.numel for b in buckets) == total
        seen = set()
        for b in buckets:
            assert b.numel_padded % (world * 64) == 0
            assert b.shard_numel * world == b.numel_padded
            for (name, p), off in zip(b.named_params, b.offsets):
                assert name not in seen
                seen.add(name)
                assert p.data.data_ptr() == b.params_flat[off:].data_ptr()
                assert torch.equal(p.data, originals[name])  # repack kept values
        assert seen == set(originals)

    check()


def test_grads_accumulate_into_flat():
    m = mnist_mlp(hidden=16)
    eng = SyncEngine(m, Comm(), bucket_cap_mb=0.01)
    x, y = synthetic_mnist_batch(8, generator=torch.Generator().manual_seed(0))
    eng.zero_grad()
    loss = m.loss(x, y)
    loss.backward()
    assert all(float(b.grads.abs().sum()) > 0 for b in eng.buckets)
    # every param's .grad is still the bucket view after backward
    for b in eng.buckets:
        for (name, p), off in zip(b.named_params, b.offsets):
            assert p.grad.data_ptr() == b.grads[off:].data_ptr(), name


def test_trajectory_matches_plain_torch():
    """world=1 engine step sequence == plain torch SGD, bitwise."""
    torch.manual_seed(0)
    m1 = mnist_mlp(hidden=32)
    m2 = mnist_mlp(hidden=32)
    m2.load_state_dict(m1.state_dict())

    eng = SyncEngine(m1, Comm(), bucket_cap_mb=0.01)
    opt1 = GradientDescentOptimizer(0.2)
    opt2 = torch.optim.SGD(m2.parameters(), lr=0.2)

    g = torch.Generator().manual_seed(1)
    for step in range(10):
        x, y = synthetic_mnist_batch(16, generator=g)
        eng.zero_grad()
        loss1 = m1.loss(x, y)
        eng.step(loss1, opt1)

        opt2.zero_grad()
        loss2 = m2.loss(x, y)
        loss2.backward()
        opt2.step()
        assert torch.equal(loss1.detach(), loss2.detach()), step
    for (n1, p1), (n2, p2) in zip(m1.named_parameters(), m2.named_parameters()):
        assert torch.allclose(p1, p2, atol=0), n1


def test_momentum_trajectory_matches_plain_torch():
    torch.manual_seed(3)
    m1 = mnist_mlp(hidden=24)
    m2 = mnist_mlp(hidden=24)
    m2.load_state_dict(m1.state_dict())
    eng = SyncEngine(m1, Comm())
    opt1 = MomentumOptimizer(0.1, 0.9)
    opt2 = torch.optim.SGD(m2.parameters(), lr=0.1, momentum=0.9)
    g = torch.Generator().manual_seed(4)
    for _ in range(6):
        x, y = synthetic_mnist_batch(16, generator=g)
        eng.zero_grad()
        eng.step(m1.loss(x, y), opt1)
        opt2.zero_grad()
        m2.loss(x, y).backward()
        opt2.step()
    for (n1, p1), (_, p2) in zip(m1.named_parameters(), m2.named_parameters()):
        assert torch.allclose(p1, p2, atol=1e-6), n1


def test_overlap_launch_order_covers_all_buckets():
    m = mnist_mlp(hidden=64)
    eng = SyncEngine(m, Comm(), bucket_cap_mb=0.02)
    x, y = synthetic_mnist_batch(4, generator=torch.Generator().manual_seed(0))
    eng.zero_grad()
    loss = m.loss(x, y)
    loss.backward()
    eng.finish_backward()
    assert len(eng._launch_order) == len(eng.buckets)


def test_resnet_tiny_cpu_smoke():
    """Bottleneck with fused bn3(add_relu) runs fwd/bwd on CPU."""
    import torch
    from rocps.models.resnet import ResNet
    torch.manual_seed(0)
    m = ResNet([1, 1, 1, 1], num_classes=10)
    x = torch.randn(2, 3, 64, 64)
    y = torch.randint(0, 10, (2,))
    loss = m.loss(x, y)
    loss.backward()
    assert torch.isfinite(loss)
    for n, p in m.named_parameters():
        assert p.grad is not None and torch.isfinite(p.grad).all(), n
