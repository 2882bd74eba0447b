"""Direct unit tests for auxiliary subsystems:
the thread Coordinator, observability (JSONL metrics, step timer,
chrome trace, event files with an independent TFRecord parse), the
async-engine shm layout/segment, and the optimizer wire spec.

These run on CPU; the end-to-end counterparts live in
test_dist_mnist.py / test_session.py.
"""

import json
import os
import struct
import threading
import time

import numpy as np
import pytest
import torch

from rocps import _bundle as C
from rocps.engine.async_engine import (_MAGIC, _ShmLayout, _ShmSeg,
                                       _build_optimizer, optimizer_spec)
from rocps.obs import EventFileWriter, MetricsLogger
from rocps.obs.metrics import StepTimer, chrome_trace
from rocps.train.coordinator import Coordinator, LooseVersionThread


# ---------------------------------------------------------------------------
# Coordinator (C14)
# ---------------------------------------------------------------------------

def test_coordinator_stop_flow():
    coord = Coordinator()
    assert not coord.should_stop()
    coord.request_stop()
    assert coord.should_stop()
    assert coord.wait_for_stop(timeout=0.1)
    coord.join([])  # no exception recorded -> returns cleanly


def test_coordinator_join_reraises_first_exception():
    coord = Coordinator()

    def boom():
        raise RuntimeError("first")

    t = LooseVersionThread(coord, boom, name="boom")
    t.start()
    # a second error later must NOT displace the first
    coord.wait_for_stop(5.0)
    coord.request_stop(ValueError("second"))
    with pytest.raises(RuntimeError, match="first"):
        coord.join(stop_grace_period_secs=5.0)


def test_coordinator_loose_thread_stops_on_request():
    coord = Coordinator()
    counter = {"n": 0}

    def tick():
        counter["n"] += 1
        time.sleep(0.005)

    t = LooseVersionThread(coord, tick)
    t.start()
    time.sleep(0.05)
    coord.request_stop()
    t.join(5.0)
    assert not t.is_alive()
    n = counter["n"]
    assert n > 0
    time.sleep(0.05)
    assert counter["n"] == n  # genuinely stopped


def test_coordinator_stop_on_exception_context():
    coord = Coordinator()
    with coord.stop_on_exception():
        raise KeyError("inside")  # swallowed, recorded
    assert coord.should_stop()
    with pytest.raises(KeyError):
        coord.raise_requested_exception()


# ---------------------------------------------------------------------------
# Observability (C16)
# ---------------------------------------------------------------------------

def test_metrics_logger_jsonl(tmp_path):
    m = MetricsLogger(str(tmp_path))
    m.log(step=1, loss=2.5)
    m.log(step=2, loss=1.25, lr=0.1)
    m.close()
    lines = (tmp_path / "metrics.jsonl").read_text().strip().split("\n")
    recs = [json.loads(ln) for ln in lines]
    assert recs[0]["step"] == 1 and recs[0]["loss"] == 2.5
    assert recs[1]["lr"] == 0.1
    assert all("t" in r for r in recs)


def test_metrics_logger_no_dir_is_noop():
    m = MetricsLogger(None)
    m.log(step=1)  # must not raise
    m.close()


def test_step_timer_cpu_wall():
    t = StepTimer(use_cuda=False)
    t.start("fwd")
    time.sleep(0.02)
    t.stop("fwd")
    t.start("fwd")
    time.sleep(0.02)
    t.stop("fwd")
    assert t.wall_ms("fwd") >= 30  # accumulated over both intervals
    assert t.device_ms("fwd") == t.wall_ms("fwd")  # CPU: same clock
    t.reset()
    assert t.wall_ms("fwd") == 0.0


def test_chrome_trace_shape(tmp_path):
    path = str(tmp_path / "trace.json")
    chrome_trace([{"name": "step", "ts": 10, "dur": 5, "tid": 1},
                  {"name": "allreduce", "ts": 12, "dur": 2}], path)
    doc = json.load(open(path))
    evs = doc["traceEvents"]
    assert [e["ph"] for e in evs] == ["X", "X"]
    assert evs[0]["name"] == "step" and evs[0]["tid"] == 1
    assert evs[1]["tid"] == 0  # default


def _parse_tfrecords(raw: bytes):
    """Independent TFRecord-framing parse: verifies both masked CRCs."""
    out, p = [], 0
    while p < len(raw):
        hdr = raw[p:p + 8]
        (length,) = struct.unpack("<Q", hdr)
        (hcrc,) = struct.unpack("<I", raw[p + 8:p + 12])
        assert C.masked_crc32c(np.frombuffer(hdr, dtype=np.uint8)) == hcrc
        payload = raw[p + 12:p + 12 + length]
        (pcrc,) = struct.unpack("<I", raw[p + 12 + length:p + 16 + length])
        assert C.masked_crc32c(np.frombuffer(payload, dtype=np.uint8)) == pcrc
        out.append(payload)
        p += 16 + length
    return out


def test_event_file_tfrecord_framing_independent_parse(tmp_path):
    w = EventFileWriter(str(tmp_path))
    w.add_scalar("loss", 3.5, step=7)
    w.close()
    raw = open(w.path, "rb").read()
    records = _parse_tfrecords(raw)
    assert len(records) == 2  # file_version event + one scalar
    # record 0: Event{wall_time=1 double, file_version=3 string}
    ev = records[0]
    assert ev[0] == (1 << 3) | 1  # field 1, 64-bit
    fv_off = 9
    assert ev[fv_off] == (3 << 3) | 2
    fv_len = ev[fv_off + 1]
    assert ev[fv_off + 2:fv_off + 2 + fv_len] == b"brain.Event:2"
    # record 1: step=7 varint, summary holds tag "loss" and value 3.5
    ev = records[1]
    assert ev[9] == (2 << 3) | 0 and ev[10] == 7
    assert b"loss" in ev
    assert struct.pack("<f", 3.5) in ev


def test_event_file_fuzz_framing():
    """Property fuzz: any sequence of scalars produces a valid TFRecord
    stream (every frame CRC-checked by the independent parser)."""
    import tempfile

    from hypothesis import given, settings, strategies as st

    @settings(deadline=None, max_examples=20)
    @given(st.lists(st.tuples(
        st.text(alphabet=st.characters(codec="utf-8",
                                       exclude_categories=("Cs",)),
                min_size=1, max_size=20),
        st.floats(allow_nan=False, allow_infinity=False, width=32),
        st.integers(0, 2**31 - 1)), min_size=0, max_size=12))
    def check(scalars):
        with tempfile.TemporaryDirectory() as d:
            w = EventFileWriter(d)
            for tag, val, step in scalars:
                w.add_scalar(tag, val, step)
            w.close()
            raw = open(w.path, "rb").read()
        records = _parse_tfrecords(raw)  # asserts every masked CRC
        assert len(records) == 1 + len(scalars)

    check()


# ---------------------------------------------------------------------------
# Async-engine shm layout (C5)
# ---------------------------------------------------------------------------

def test_shm_layout_alignment_and_disjointness():
    lay = _ShmLayout(numel=1000, num_workers=3, itemsize=4)
    assert lay.header_words == 8 + 9
    assert lay.params_off % 256 == 0
    assert lay.params_off >= lay.header_words * 8
    grad_bytes = 1000 * 4
    spans = [(0, lay.header_words * 8),
             (lay.params_off, lay.params_off + grad_bytes)]
    for w in range(3):
        off = lay.grad_off(w)
        assert off % 256 == 0
        spans.append((off, off + grad_bytes))
    spans.sort()
    for (a0, a1), (b0, b1) in zip(spans, spans[1:]):
        assert a1 <= b0, "overlapping regions"
    assert lay.total >= spans[-1][1]
    # per-worker header slots are distinct and inside the header
    idxs = [f(w) for w in range(3)
            for f in (_ShmLayout.seq_idx, _ShmLayout.ack_idx,
                      _ShmLayout.pver_idx)]
    assert len(set(idxs)) == 9 and max(idxs) < lay.header_words


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_shm_seg_cross_mapping_roundtrip(tmp_path, dtype):
    """Writer mapping and a second (attach) mapping see the same bytes —
    the PS/worker shared-memory contract, minus the /dev/shm path."""
    lay = _ShmLayout(numel=64, num_workers=2, itemsize=2 if dtype == torch.bfloat16 else 4)
    path = str(tmp_path / "seg")
    a = _ShmSeg(path, lay, dtype, create=True)
    b = _ShmSeg(path, lay, dtype, create=False)
    try:
        a.header[0] = _MAGIC
        a.header[_ShmLayout.seq_idx(1)] = 41
        src = torch.linspace(-2, 2, 64).to(dtype)
        a.params.copy_(src)
        a.grads[1].copy_(src * 2)
        assert b.header[0].item() == _MAGIC
        assert b.header[_ShmLayout.seq_idx(1)].item() == 41
        assert torch.equal(b.params, src)
        assert torch.equal(b.grads[1], src * 2)
        assert b.params.dtype == dtype
        # reverse direction: attach side acks, creator observes
        b.header[_ShmLayout.ack_idx(1)] = 41
        assert a.header[_ShmLayout.ack_idx(1)].item() == 41
    finally:
        a.close()
        b.close()


# ---------------------------------------------------------------------------
# Comm world_size==1 fast path (C6)
# ---------------------------------------------------------------------------

def test_comm_world1_shortcircuits():
    from rocps.engine.comm import Comm
    c = Comm(rank=0, world_size=1, backend="gloo",
             device=torch.device("cpu"))
    assert not c.is_distributed
    t = torch.arange(4.0)
    assert c.all_reduce(t).wait()
    out = torch.zeros(4)
    c.reduce_scatter(out, t.clone()).wait()
    assert torch.equal(out, t)
    gathered = torch.zeros(4)
    c.all_gather(gathered, out).wait()
    assert torch.equal(gathered, t)
    c.broadcast(t).wait()
    c.barrier()
    c.destroy()  # no group -> no-op


def test_comm_tune_env(monkeypatch):
    from rocps.engine.comm import Comm
    monkeypatch.delenv("NCCL_MIN_NCHANNELS", raising=False)
    monkeypatch.delenv("ROCPS_DETERMINISTIC", raising=False)
    monkeypatch.setenv("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    Comm.tune_env()
    # >=8 channels so rings stripe across the 7 xGMI links
    assert int(os.environ["NCCL_MIN_NCHANNELS"]) >= 8
    assert os.environ["HSA_ENABLE_IPC_MODE_LEGACY"] == "0"


def test_comm_from_env_single(monkeypatch):
    """bench.py single-process path: no WORLD_SIZE -> world-1 comm with
    no process group."""
    from rocps.engine.comm import Comm
    for k in ("WORLD_SIZE", "RANK", "LOCAL_RANK"):
        monkeypatch.delenv(k, raising=False)
    c = Comm.from_env()
    assert c.world_size == 1 and c.rank == 0
    assert not c.is_distributed


def test_deterministic_mode(monkeypatch):
    """ROCPS_DETERMINISTIC=1: one RCCL channel + fixed schedule, and
    the sync engine serializes bucket launches (overlap off)."""
    from rocps.engine.comm import Comm
    from rocps.engine.sync_engine import SyncEngine
    from rocps.models import mnist_mlp

    monkeypatch.setenv("ROCPS_DETERMINISTIC", "1")
    for k in ("NCCL_MIN_NCHANNELS", "NCCL_MAX_NCHANNELS",
              "NCCL_ALGO", "NCCL_PROTO"):
        monkeypatch.delenv(k, raising=False)
    Comm.tune_env()
    assert os.environ["NCCL_MAX_NCHANNELS"] == "1"
    assert os.environ["NCCL_ALGO"] == "Ring"
    eng = SyncEngine(mnist_mlp(hidden=8),
                     Comm(device=torch.device("cpu"), backend="gloo"),
                     mode="allreduce", bucket_cap_mb=0.01)
    assert eng.overlap is False
    eng.detach()


# ---------------------------------------------------------------------------
# Optimizer wire spec (worker -> PS handshake)
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("make", [
    lambda: __import__("rocps.train", fromlist=["x"]).GradientDescentOptimizer(0.05),
    lambda: __import__("rocps.train", fromlist=["x"]).MomentumOptimizer(0.2, 0.95, use_nesterov=True),
    lambda: __import__("rocps.train", fromlist=["x"]).AdamOptimizer(3e-4, 0.8, 0.95, 1e-6),
])
def test_optimizer_spec_roundtrip(make):
    opt = make()
    spec = optimizer_spec(opt)
    rebuilt = _build_optimizer(spec)
    assert optimizer_spec(rebuilt) == spec
    # the rebuilt optimizer steps a parameter identically to the original
    p1 = torch.full((8,), 1.0)
    p2 = p1.clone()
    g = torch.linspace(0.1, 0.8, 8)
    for o, p in ((opt, p1), (rebuilt, p2)):
        o.apply_flat(p, g.clone(), slot=0)
    assert torch.equal(p1, p2)


def test_queue_runner_prefetch_and_stop():
    """C14 input feeding: producer threads fill a bounded queue; the
    consumer sees every index exactly once (per-runner counter);
    coordinator stop drains cleanly."""
    import torch

    from rocps.train.coordinator import Coordinator
    from rocps.train.queue_runner import QueueRunner, start_queue_runners

    def batch_fn(i):
        g = torch.Generator().manual_seed(i)
        return i, torch.randn(4, generator=g)

    qr = QueueRunner(batch_fn, capacity=3, num_threads=2)
    coord = start_queue_runners([qr])
    seen = set()
    for _ in range(20):
        i, t = qr.dequeue(timeout=10)
        assert i not in seen
        seen.add(i)
        g = torch.Generator().manual_seed(i)
        assert torch.equal(t, torch.randn(4, generator=g))
    coord.request_stop()
    coord.join()


def test_queue_runner_producer_exception_routes_to_coordinator():
    import pytest

    from rocps.train.queue_runner import QueueRunner, start_queue_runners

    def bad_fn(i):
        if i >= 3:
            raise ValueError("boom")
        return i

    qr = QueueRunner(bad_fn, capacity=2, num_threads=1)
    coord = start_queue_runners([qr])
    got = []
    with pytest.raises((RuntimeError, TimeoutError)):
        for _ in range(10):
            got.append(qr.dequeue(timeout=5))
    assert coord.should_stop()
    assert got == [0, 1, 2] or len(got) <= 3


def test_elastic_recovery_rejects_non_sync_trainer():
    """ElasticRecovery is a collective-engine feature; a semisync
    trainer (mode routed to the PS engine) must be rejected loudly."""
    import pytest

    from rocps.train.recovery import ElasticRecovery

    class FakeTrainer:
        mode = "async"

    with pytest.raises(ValueError):
        ElasticRecovery(FakeTrainer(), None)
