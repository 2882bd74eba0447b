"""MonitoredTrainingSession + hooks + Saver tests (C12/C13)."""

import os

import pytest
import torch

from rocps.models import mnist_mlp, synthetic_mnist_batch
from rocps.train import (AdamOptimizer, GradientDescentOptimizer,
                         MomentumOptimizer, MonitoredTrainingSession,
                         NanTensorHook, Saver, StepCounterHook, StopAtStepHook,
                         Trainer, latest_checkpoint)


def _run(trainer, steps, ckdir=None, seed=5, save_every=None, hooks=()):
    g = torch.Generator().manual_seed(seed)
    losses = []
    with MonitoredTrainingSession(
            trainer, checkpoint_dir=ckdir,
            save_checkpoint_steps=save_every,
            save_checkpoint_secs=None if save_every else 600,
            hooks=[StopAtStepHook(last_step=steps), *hooks]) as sess:
        while not sess.should_stop():
            gg = torch.Generator().manual_seed(seed + trainer.global_step)
            x, y = synthetic_mnist_batch(16, generator=gg)
            losses.append(float(sess.run(lambda: trainer.model.loss(x, y))))
    return losses


def test_stop_at_step():
    tr = Trainer(mnist_mlp(hidden=8), GradientDescentOptimizer(0.1))
    losses = _run(tr, 7)
    assert tr.global_step == 7 and len(losses) == 7


def test_nan_hook_raises():
    class BadModel(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.w = torch.nn.Parameter(torch.ones(1))

        def loss(self, x, y):
            return self.w.sum() * float("nan")

    tr = Trainer(BadModel(), GradientDescentOptimizer(0.1))
    with pytest.raises(RuntimeError, match="NaN"):
        _run(tr, 3, hooks=[NanTensorHook()])


def test_checkpoint_resume_exact_continuation(tmp_path):
    """Train 30 straight vs train 15 + resume + 15: identical params."""
    ckdir = str(tmp_path / "ck")
    torch.manual_seed(11)
    mA = mnist_mlp(hidden=16)
    trA = Trainer(mA, MomentumOptimizer(0.1, 0.9))
    _run(trA, 30)

    torch.manual_seed(11)
    mB = mnist_mlp(hidden=16)
    trB = Trainer(mB, MomentumOptimizer(0.1, 0.9))
    _run(trB, 15, ckdir=ckdir, save_every=15)
    assert latest_checkpoint(ckdir).endswith("-15")

    torch.manual_seed(99)  # different init — restore must overwrite it
    mC = mnist_mlp(hidden=16)
    trC = Trainer(mC, MomentumOptimizer(0.1, 0.9))
    _run(trC, 30, ckdir=ckdir, save_every=15)  # restores at 15, runs to 30
    assert trC.global_step == 30
    pA = dict(mA.named_parameters())
    pC = dict(mC.named_parameters())
    for k in pA:
        assert torch.allclose(pA[k], pC[k], atol=1e-7), k


def test_checkpoint_resume_exact_adam(tmp_path):
    """Adam resume must also restore the bias-correction step counter
    (t): with t reset, the resumed alpha_t is ~2x too large and the
    trajectories diverge."""
    from rocps.train import AdamOptimizer

    ckdir = str(tmp_path / "cka")
    torch.manual_seed(11)
    mA = mnist_mlp(hidden=16)
    trA = Trainer(mA, AdamOptimizer(0.01))
    _run(trA, 30)

    torch.manual_seed(11)
    mB = mnist_mlp(hidden=16)
    trB = Trainer(mB, AdamOptimizer(0.01))
    _run(trB, 15, ckdir=ckdir, save_every=15)

    torch.manual_seed(99)
    mC = mnist_mlp(hidden=16)
    trC = Trainer(mC, AdamOptimizer(0.01))
    _run(trC, 30, ckdir=ckdir, save_every=15)
    assert trC.global_step == 30
    pA = dict(mA.named_parameters())
    pC = dict(mC.named_parameters())
    for k in pA:
        assert torch.allclose(pA[k], pC[k], atol=1e-6), k


def test_checkpoint_includes_optimizer_state(tmp_path):
    ckdir = str(tmp_path / "ck2")
    tr = Trainer(mnist_mlp(hidden=8), AdamOptimizer(1e-3))
    _run(tr, 5, ckdir=ckdir, save_every=5)
    from rocps.checkpoint.bundle import list_bundle
    names = {e["name"] for e in list_bundle(latest_checkpoint(ckdir))}
    assert "global_step" in names
    assert any(n.startswith("optimizer/") and n.endswith("/m") for n in names)
    assert any(n.startswith("optimizer/") and n.endswith("/v") for n in names)


def test_max_to_keep_gc(tmp_path):
    ckdir = str(tmp_path / "ck3")
    tr = Trainer(mnist_mlp(hidden=8), GradientDescentOptimizer(0.1))
    saver = Saver(tr, max_to_keep=2)
    for s in [1, 2, 3]:
        tr.global_step = s
        saver.save(None, os.path.join(ckdir, "model.ckpt"), global_step=s)
    files = os.listdir(ckdir)
    assert "model.ckpt-1.index" not in files
    assert "model.ckpt-2.index" in files and "model.ckpt-3.index" in files
    assert latest_checkpoint(ckdir).endswith("model.ckpt-3")
    assert Saver.last_checkpoints(ckdir) == ["model.ckpt-2", "model.ckpt-3"]


def test_saver_raw_module_source(tmp_path):
    """Saver accepts a bare nn.Module (state_dict save / load_state_dict
    restore), not just a Trainer."""
    torch.manual_seed(1)
    m = mnist_mlp(hidden=8)
    s = Saver(m)
    prefix = s.save(None, str(tmp_path / "raw" / "model.ckpt"),
                    global_step=3)
    assert prefix.endswith("-3")
    torch.manual_seed(2)
    m2 = mnist_mlp(hidden=8)
    assert not all(torch.equal(a, b) for (_, a), (_, b)
                   in zip(m.named_parameters(), m2.named_parameters()))
    Saver(m2).restore(None, prefix)
    for (n, a), (_, b) in zip(m.named_parameters(), m2.named_parameters()):
        assert torch.equal(a, b), n


def test_keep_checkpoint_every_n_hours(tmp_path):
    """n=0: every GC'd checkpoint is preserved on disk (off the state
    file); the default (10000h) deletes as usual — reference Saver
    semantics."""
    ckdir = str(tmp_path / "ck4")
    tr = Trainer(mnist_mlp(hidden=8), GradientDescentOptimizer(0.1))
    saver = Saver(tr, max_to_keep=1, keep_checkpoint_every_n_hours=0.0)
    for s in [1, 2, 3]:
        tr.global_step = s
        saver.save(None, os.path.join(ckdir, "model.ckpt"), global_step=s)
    files = os.listdir(ckdir)
    # dropped from rotation but preserved on disk
    assert "model.ckpt-1.index" in files and "model.ckpt-2.index" in files
    assert Saver.last_checkpoints(ckdir) == ["model.ckpt-3"]
    assert latest_checkpoint(ckdir).endswith("model.ckpt-3")


def test_summary_and_logging_hooks(tmp_path, capsys):
    import json as _json

    from rocps.obs import MetricsLogger
    from rocps.train.hooks import LoggingTensorHook, SummarySaverHook

    tr = Trainer(mnist_mlp(hidden=8), GradientDescentOptimizer(0.1))
    metrics = MetricsLogger(str(tmp_path))
    _run(tr, 6, hooks=[SummarySaverHook(metrics, every_n_steps=2),
                       LoggingTensorHook(every_n_iter=3)])
    metrics.close()
    recs = [_json.loads(ln) for ln in
            (tmp_path / "metrics.jsonl").read_text().strip().split("\n")]
    assert [r["step"] for r in recs] == [2, 4, 6]
    assert all("loss" in r for r in recs)
    out = capsys.readouterr().out
    assert "[rocps] step 3 loss" in out and "[rocps] step 6 loss" in out


def test_stop_at_step_num_steps_relative():
    """num_steps counts from the session-start step (restore-aware)."""
    import pytest as _pytest

    from rocps.train.hooks import StopAtStepHook

    tr = Trainer(mnist_mlp(hidden=8), GradientDescentOptimizer(0.1))
    tr.global_step = 10
    _run(tr, 100, hooks=[StopAtStepHook(num_steps=5)])
    assert tr.global_step == 15
    with _pytest.raises(ValueError):
        StopAtStepHook()
    with _pytest.raises(ValueError):
        StopAtStepHook(num_steps=1, last_step=2)


def test_profiler_hook(tmp_path):
    import json as _json

    from rocps.train.hooks import ProfilerHook

    tr = Trainer(mnist_mlp(hidden=8), GradientDescentOptimizer(0.1))
    pdir = str(tmp_path / "prof")
    _run(tr, 7, hooks=[ProfilerHook(save_steps=3, output_dir=pdir)])
    files = sorted(os.listdir(pdir))
    assert "trace_step3.json" in files and "trace_step6.json" in files
    doc = _json.load(open(os.path.join(pdir, "trace_step3.json")))
    assert doc["traceEvents"]  # valid chrome trace with events


def test_step_counter_hook():
    tr = Trainer(mnist_mlp(hidden=8), GradientDescentOptimizer(0.1))
    sc = StepCounterHook(every_n_steps=2, batch_size=16)
    _run(tr, 6, hooks=[sc])
    assert sc.last_steps_per_sec is not None and sc.last_steps_per_sec > 0


def test_supervisor_shim(tmp_path):
    import rocps
    tr = Trainer(mnist_mlp(hidden=8), GradientDescentOptimizer(0.1))
    sv = rocps.train.Supervisor(is_chief=True, logdir=str(tmp_path / "sv"),
                                trainer=tr, save_model_secs=0)
    g = torch.Generator().manual_seed(0)
    with sv.managed_session() as sess:
        for _ in range(3):
            x, y = synthetic_mnist_batch(8, generator=g)
            sess.run(lambda: tr.model.loss(x, y))
    assert tr.global_step == 3


def test_event_file_writer(tmp_path):
    import glob
    import struct

    import numpy as np

    from rocps import _bundle as C
    from rocps.obs import EventFileWriter
    w = EventFileWriter(str(tmp_path / "ev"))
    for s in range(5):
        w.add_scalar("loss", 2.0 - 0.1 * s, s)
    w.close()
    files = glob.glob(str(tmp_path / "ev" / "events.out.tfevents.*"))
    assert len(files) == 1
    raw = open(files[0], "rb").read()
    off, n = 0, 0
    while off < len(raw):
        (ln,) = struct.unpack("<Q", raw[off:off + 8])
        (lc,) = struct.unpack("<I", raw[off + 8:off + 12])
        assert lc == C.masked_crc32c(
            np.frombuffer(raw[off:off + 8], dtype=np.uint8))
        data = raw[off + 12:off + 12 + ln]
        (dc,) = struct.unpack("<I", raw[off + 12 + ln:off + 16 + ln])
        assert dc == C.masked_crc32c(np.frombuffer(data, dtype=np.uint8))
        off += 16 + ln
        n += 1
    assert n == 6  # file_version + 5 scalars


def test_resnet_tiny_checkpoint_roundtrip(tmp_path):
    """ResNet checkpoint: conv weights + BN params/buffers (incl. the
    int64 num_batches_tracked) through the TF-V2 bundle."""
    from rocps.models.resnet import ResNet
    torch.manual_seed(21)
    m = ResNet([1, 1, 1, 1], num_classes=10)
    tr = Trainer(m, MomentumOptimizer(0.05, 0.9))
    g = torch.Generator().manual_seed(0)
    for _ in range(2):
        x = torch.randn(2, 3, 64, 64, generator=g)
        y = torch.randint(0, 10, (2,), generator=g)
        tr.train_step(lambda: m.loss(x, y))
    saver = Saver(tr)
    prefix = saver.save(None, str(tmp_path / "r.ckpt"), global_step=2)

    torch.manual_seed(99)
    m2 = ResNet([1, 1, 1, 1], num_classes=10)
    tr2 = Trainer(m2, MomentumOptimizer(0.05, 0.9))
    Saver(tr2).restore(tr2, prefix)
    assert tr2.global_step == 2
    sd1, sd2 = m.state_dict(), m2.state_dict()
    for k in sd1:
        assert torch.equal(sd1[k], sd2[k]), k


def test_new_checkpoint_reader(tmp_path):
    import rocps
    from rocps.checkpoint.bundle import write_bundle
    prefix = str(tmp_path / "m.ckpt-5")
    t = {"a/w": torch.randn(3, 4), "b": torch.arange(6, dtype=torch.int64)}
    write_bundle(prefix, t)
    r = rocps.train.NewCheckpointReader(prefix)
    assert r.get_variable_to_shape_map() == {"a/w": [3, 4], "b": [6]}
    assert r.has_tensor("a/w") and not r.has_tensor("zzz")
    assert torch.equal(r.get_tensor("b"), t["b"])
    assert torch.equal(r.get_tensor("a/w"), t["a/w"])


def test_lr_schedules_and_hook():
    import rocps
    lr = rocps.train.exponential_decay(0.1, 10, 0.5, staircase=True)
    assert lr(0) == 0.1 and lr(10) == 0.05 and lr(25) == 0.025
    lr2 = rocps.train.cosine_decay(1.0, 100)
    assert abs(lr2(0) - 1.0) < 1e-9 and abs(lr2(100)) < 1e-9
    lr3 = rocps.train.polynomial_decay(1.0, 100, end_learning_rate=0.1)
    assert abs(lr3(100) - 0.1) < 1e-9

    tr = Trainer(mnist_mlp(hidden=8), GradientDescentOptimizer(0.1))
    hook = rocps.train.LearningRateHook(
        tr.optimizer, rocps.train.exponential_decay(0.1, 2, 0.5,
                                                    staircase=True))
    _run(tr, 5, hooks=[hook])
    assert abs(tr.optimizer.lr - 0.1 * 0.5 ** 2) < 1e-12
