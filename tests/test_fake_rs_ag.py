"""rs_ag engine path (reduce-scatter + shard optimizer + all-gather —
the RCCL recast of the reference's PS sync mode, C4) validated on CPU
through the in-process fake collective backend."""

import threading

import pytest
import torch

from rocps.engine.fake_comm import FakeComm, FakeWorld, run_fake_cluster
from rocps.engine.sync_engine import SyncEngine
from rocps.models import mnist_mlp, synthetic_mnist_batch
from rocps.train.optimizer import GradientDescentOptimizer, MomentumOptimizer


def _ref_run(steps, global_batch, lr, momentum=None, seed=3):
    torch.manual_seed(7)
    m = mnist_mlp(hidden=32)
    opt = (torch.optim.SGD(m.parameters(), lr=lr, momentum=momentum)
           if momentum else torch.optim.SGD(m.parameters(), lr=lr))
    losses = []
    for s in range(steps):
        g = torch.Generator().manual_seed(seed + s)
        x, y = synthetic_mnist_batch(global_batch, generator=g)
        opt.zero_grad()
        loss = m.loss(x, y)
        loss.backward()
        opt.step()
        losses.append(float(loss.detach()))
    return m, losses


def _fake_run(world, steps, per_rank_batch, lr, momentum=None, mode="rs_ag",
              seed=3):
    results = {}
    lock = threading.Lock()
    # build models on the MAIN thread: torch.manual_seed is process-global,
    # seeding from rank threads races the RNG draws of model init
    models = []
    for _ in range(world):
        torch.manual_seed(7)
        models.append(mnist_mlp(hidden=32))

    def rank_fn(comm: FakeComm, rank: int):
        m = models[rank]
        eng = SyncEngine(m, comm, mode=mode, bucket_cap_mb=0.01)
        opt = (MomentumOptimizer(lr, momentum) if momentum
               else GradientDescentOptimizer(lr))
        eng.broadcast_parameters(src=0)
        losses = []
        for s in range(steps):
            g = torch.Generator().manual_seed(seed + s)
            x, y = synthetic_mnist_batch(per_rank_batch * world,
                                         generator=g)
            lo = rank * per_rank_batch
            xb, yb = x[lo:lo + per_rank_batch], y[lo:lo + per_rank_batch]
            eng.zero_grad()
            loss = m.loss(xb, yb)
            eng.step(loss, opt)
            losses.append(float(loss.detach()))
        with lock:
            results[rank] = (m, losses)
        return rank

    run_fake_cluster(world, rank_fn)
    return results


@pytest.mark.parametrize("mode", ["rs_ag", "allreduce"])
def test_fake_two_ranks_match_single_process(mode):
    steps, B, lr = 8, 16, 0.2
    ref_model, _ = _ref_run(steps, 2 * B, lr)
    results = _fake_run(2, steps, B, lr, mode=mode)
    p_ref = dict(ref_model.named_parameters())
    for rank in (0, 1):
        m, _ = results[rank]
        for n, p in m.named_parameters():
            assert torch.allclose(p, p_ref[n], atol=1e-5), (mode, rank, n)
    # both ranks end with identical replicas
    m0, m1 = results[0][0], results[1][0]
    for (n, a), (_, b) in zip(m0.named_parameters(), m1.named_parameters()):
        assert torch.equal(a, b), n


def test_fake_four_ranks_momentum_rs_ag():
    steps, B, lr, mu = 6, 8, 0.1, 0.9
    ref_model, _ = _ref_run(steps, 4 * B, lr, momentum=mu)
    results = _fake_run(4, steps, B, lr, momentum=mu, mode="rs_ag")
    p_ref = dict(ref_model.named_parameters())
    m, _ = results[0]
    for n, p in m.named_parameters():
        assert torch.allclose(p, p_ref[n], atol=1e-5), n


def test_fake_bn_buffer_sync():
    """BN running stats are rank-local between saves; state collection
    averages them across replicas (the reference keeps moving stats as
    shared PS variables)."""
    from rocps.train.session import Trainer

    class TinyBN(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.fc = torch.nn.Linear(4, 4)
            self.bn = torch.nn.BatchNorm1d(4)

        def loss(self, x, y):
            return ((self.bn(self.fc(x)) - y) ** 2).mean()

    models = []
    for _ in range(2):
        torch.manual_seed(7)
        models.append(TinyBN())
    results = {}
    lock = threading.Lock()

    def rank_fn(comm: FakeComm, rank: int):
        tr = Trainer(models[rank], GradientDescentOptimizer(0.01),
                     comm=comm, sync_mode="rs_ag", bucket_cap_mb=0.01)
        g = torch.Generator().manual_seed(100 + rank)  # DIFFERENT data
        x = torch.randn(8, 4, generator=g) * (1.0 + rank)
        y = torch.zeros(8, 4)
        for _ in range(3):
            tr.train_step(lambda: tr.model.loss(x, y))
        local_mean = tr.model.bn.running_mean.clone()
        st = tr.state_tensors()  # collective: averages buffers
        with lock:
            results[rank] = (local_mean, st,
                             tr.model.bn.running_mean.clone())
        return rank

    run_fake_cluster(2, rank_fn)
    m0, st0, after0 = results[0]
    m1, st1, after1 = results[1]
    assert not torch.allclose(m0, m1)  # drifted apart (different data)
    expect = 0.5 * (m0 + m1)
    assert torch.allclose(st0["bn.running_mean"], expect, atol=1e-6)
    assert torch.equal(after0, after1)  # replicas re-synced in place


@pytest.mark.parametrize("world", [2, 8])
def test_fake_rs_ag_checkpoint_save_restore_sharded_state(world):
    """Trainer-level save/restore over the rs_ag engine: collecting the
    checkpoint all-gathers each rank's optimizer-state shard and restore
    re-scatters it — the exact path the RCCL engine takes on GPU.
    12 straight steps == 6 steps + save + restore-into-fresh-world + 6.
    world=8 is the MI355X node shape (maximize 8-rank
    readiness evidence without an 8-GPU node)."""
    from rocps.train.session import Trainer

    steps, B, lr, mu = 6, 8, 0.1, 0.9

    def trainer_run(models, steps, start_step, state=None, out_state=None):
        results = {}
        lock = threading.Lock()

        def rank_fn(comm: FakeComm, rank: int):
            tr = Trainer(models[rank], MomentumOptimizer(lr, mu),
                         comm=comm, sync_mode="rs_ag", bucket_cap_mb=0.01)
            tr.global_step = start_step
            if state is not None:
                tr.load_state_tensors(state)
            tr.engine.broadcast_parameters(src=0)
            for s in range(tr.global_step, tr.global_step + steps):
                g = torch.Generator().manual_seed(3 + s)
                x, y = synthetic_mnist_batch(world * B, generator=g)
                lo = rank * B
                xb, yb = x[lo:lo + B], y[lo:lo + B]
                tr.train_step(lambda: tr.model.loss(xb, yb))
            st = tr.state_tensors()  # collective on every rank
            with lock:
                results[rank] = st
            return rank

        run_fake_cluster(world, rank_fn)
        if out_state is not None:
            out_state.update(results[0])
        return results

    # straight 12-step reference
    models = []
    for _ in range(world):
        torch.manual_seed(7)
        models.append(mnist_mlp(hidden=32))
    straight = trainer_run(models, 12, 0)

    # 6 steps, checkpoint, restore into a FRESH world, 6 more
    models = []
    for _ in range(world):
        torch.manual_seed(7)
        models.append(mnist_mlp(hidden=32))
    saved = {}
    trainer_run(models, steps, 0, out_state=saved)
    assert saved["global_step"].item() == 6
    assert any(k.startswith("optimizer/bucket") for k in saved)

    models = []
    for _ in range(world):
        torch.manual_seed(99)  # different init; restore must overwrite
        models.append(mnist_mlp(hidden=32))
    resumed = trainer_run(models, steps, 6, state=saved)

    for k, v in straight[0].items():
        assert torch.allclose(v.float(), resumed[0][k].float(),
                              atol=1e-5), k
    def rank_fn(comm: FakeComm, rank: int):
        t = torch.full((4,), float(rank + 1))
        comm.all_reduce(t)
        assert torch.equal(t, torch.full((4,), 6.0))  # 1+2+3
        inp = torch.arange(6, dtype=torch.float32) + 10 * rank
        out = torch.empty(2)
        comm.reduce_scatter(out, inp)
        expect = (torch.arange(6, dtype=torch.float32) * 3 + 30)[
            rank * 2:rank * 2 + 2]
        assert torch.equal(out, expect)
        full = torch.empty(6)
        comm.all_gather(full, out)
        b = torch.full((2,), float(rank))
        comm.broadcast(b, src=1)
        assert torch.equal(b, torch.full((2,), 1.0))
        return True

    assert run_fake_cluster(3, rank_fn) == [True, True, True]


def test_fake_fault_injection_surfaces_error():
    def rank_fn(comm_args):
        pass

    world = FakeWorld(2, timeout=5.0)
    errors = []

    def runner(rank):
        comm = FakeComm(world, rank, fail_at=3 if rank == 1 else None)
        try:
            for _ in range(5):
                t = torch.ones(2)
                comm.all_reduce(t)
        except BaseException as e:  # noqa: BLE001
            errors.append((rank, type(e).__name__))
            world.barrier.abort()

    ts = [threading.Thread(target=runner, args=(r,)) for r in range(2)]
    for t in ts:
        t.start()
    for t in ts:
        t.join(20)
    kinds = {k for _, k in errors}
    assert "ConnectionError" in kinds  # the injected failure
    assert len(errors) == 2  # the peer observes a broken collective


def test_fake_rs_ag_bf16_master_weights():
    """The GPU bench configuration exactly: bf16 params + bf16 grads,
    fp32 master weights updated on the reduce-scattered shard."""
    world, steps, B, lr = 2, 6, 16, 0.1
    models = []
    for _ in range(world):
        torch.manual_seed(11)
        models.append(mnist_mlp(hidden=32).bfloat16())
    # fp32 reference with the same bf16 forward/grad dtypes is awkward;
    # instead check cross-rank bitwise agreement + fp32-master tracking.
    results = {}

    def rank_fn(comm, rank):
        m = models[rank]
        eng = SyncEngine(m, comm, mode="rs_ag", bucket_cap_mb=0.01)
        opt = GradientDescentOptimizer(lr)
        eng.broadcast_parameters(src=0)
        for s in range(steps):
            g = torch.Generator().manual_seed(100 + s)
            x, y = synthetic_mnist_batch(world * B, generator=g)
            lo = rank * B
            eng.zero_grad()
            loss = m.loss(x[lo:lo + B].bfloat16(), y[lo:lo + B])
            eng.step(loss, opt)
        results[rank] = (m, opt)
        return rank

    run_fake_cluster(world, rank_fn)
    m0, opt0 = results[0]
    m1, opt1 = results[1]
    for (n, a), (_, b) in zip(m0.named_parameters(), m1.named_parameters()):
        assert a.dtype == torch.bfloat16
        assert torch.equal(a, b), n  # replicas bitwise identical
    # every bucket created fp32 master state on its shard
    eng_slots = [k for k in opt0._state if isinstance(k, tuple)]
    assert eng_slots
    for k in eng_slots:
        st = opt0._state[k]
        assert "master" in st and st["master"].dtype == torch.float32
