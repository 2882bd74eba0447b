"""ClusterSpec + control-plane tests (capability C1)."""

import threading
import time

import pytest

from rocps.cluster import ClusterSpec, ControlClient, Server


def test_clusterspec_api():
    cs = ClusterSpec({"ps": ["h0:2222"], "worker": ["h1:2223", "h2:2224"]})
    assert cs.jobs == ["ps", "worker"]
    assert cs.num_tasks("worker") == 2
    assert cs.task_address("worker", 1) == "h2:2224"
    assert cs.job_tasks("ps") == ["h0:2222"]
    assert cs.as_dict() == {"ps": ["h0:2222"], "worker": ["h1:2223", "h2:2224"]}
    assert cs.master_job == "ps"
    assert cs.master_address == "h0:2222"
    assert ClusterSpec(cs) == cs
    # dict-of-dict form (sparse task indices, like TF)
    cs2 = ClusterSpec({"worker": {0: "a:1", 2: "b:2"}})
    assert cs2.task_indices("worker") == [0, 2]
    assert cs2.master_job == "worker"


def _mk_cluster(ports):
    return ClusterSpec({"ps": [f"127.0.0.1:{ports[0]}"],
                        "worker": [f"127.0.0.1:{ports[1]}",
                                   f"127.0.0.1:{ports[2]}"]})


def test_control_plane_kv_counter_barrier(free_ports):
    ports = free_ports(3)
    cs = _mk_cluster(ports)
    master = Server(cs, "ps", 0)
    try:
        c = ControlClient(cs.master_address)
        c.put("alpha", {"x": 1})
        assert c.get("alpha") == {"x": 1}
        with pytest.raises(KeyError):
            c.get("missing", blocking=False)
        assert c.incr("ctr") == 1
        assert c.incr("ctr", 5) == 6
        assert c.read_counter("ctr") == 6
        c.put_counter("ctr", 100)
        assert c.read_counter("ctr") == 100

        # blocking get satisfied by a later put
        got = {}

        def getter():
            got["v"] = ControlClient(cs.master_address).get(
                "later", blocking=True, timeout=10)

        t = threading.Thread(target=getter)
        t.start()
        time.sleep(0.1)
        c.put("later", 42)
        t.join(5)
        assert got["v"] == 42

        # 3-party barrier
        results = []

        def arrive(who):
            cc = ControlClient(cs.master_address)
            cc.barrier("b1", who, 3, timeout=10)
            results.append(who)

        ts = [threading.Thread(target=arrive, args=(f"w{i}",)) for i in range(3)]
        for t in ts:
            t.start()
        for t in ts:
            t.join(10)
        assert len(results) == 3

        # registration / members
        c.register("worker", 0)
        assert "worker:0" in c.members()
        c.close()
    finally:
        master.stop()


def test_server_join_released_by_shutdown(free_ports):
    ports = free_ports(3)
    cs = _mk_cluster(ports)
    srv = Server(cs, "ps", 0)
    done = []

    def joiner():
        srv.join()
        done.append(True)

    t = threading.Thread(target=joiner)
    t.start()
    time.sleep(0.1)
    assert not done
    ControlClient(cs.master_address).shutdown_cluster()
    t.join(5)
    assert done
    srv.stop()


def test_worker_server_dispatch(free_ports):
    ports = free_ports(3)
    cs = _mk_cluster(ports)
    master = Server(cs, "ps", 0)
    worker = Server(cs, "worker", 1)
    try:
        c = ControlClient(cs.task_address("worker", 1))
        r = c.call("ping")
        assert r["ok"] and r["job"] == "worker" and r["task"] == 1
        # custom handler with binary payload round-trip
        worker.register_handler(
            "echo", lambda head, payload: ({"ok": True, "n": head["n"]},
                                           payload[::-1]))
        r, pl = c.call_raw("echo", payload=b"abcdef", n=7)
        assert r["n"] == 7 and pl == b"fedcba"
        c.close()
    finally:
        master.stop()
        worker.stop()


def test_create_local_cluster():
    """tf.test.create_local_cluster parity: in-process multi-server."""
    from rocps.testing import create_local_cluster
    workers, ps, cluster = create_local_cluster(num_workers=2, num_ps=1)
    try:
        assert cluster.num_tasks("worker") == 2
        assert cluster.num_tasks("ps") == 1
        c = ControlClient(cluster.master_address)
        assert c.incr("k") == 1
        for s in workers:
            cc = ControlClient(f"127.0.0.1:{s.bound_port}")
            assert cc.call("ping")["ok"]
            cc.close()
        c.close()
    finally:
        for s in workers + ps:
            s.stop()


def test_control_plane_concurrency_stress(free_ports):
    """Hammer the native server's kv/counter/barrier state from many
    threads — catches data races in csrc/control/control_plane.cc."""
    import threading
    ports = free_ports(3)
    cs = _mk_cluster(ports)
    master = Server(cs, "ps", 0)
    try:
        n_threads, n_ops = 16, 60
        errors = []

        def worker(t):
            try:
                c = ControlClient(cs.master_address)
                for i in range(n_ops):
                    c.incr("shared")
                    c.put(f"k{t}_{i}", {"t": t, "i": i})
                    assert c.get(f"k{t}_{i}") == {"t": t, "i": i}
                c.barrier("stress", f"t{t}", n_threads, timeout=60)
                c.close()
            except Exception as e:  # noqa: BLE001
                errors.append(e)

        ts = [threading.Thread(target=worker, args=(t,))
              for t in range(n_threads)]
        for t in ts:
            t.start()
        for t in ts:
            t.join(90)
        assert not errors, errors[:3]
        c = ControlClient(cs.master_address)
        assert c.read_counter("shared") == n_threads * n_ops
        c.close()
    finally:
        master.stop()


def test_pure_python_server_fallback(free_ports, monkeypatch):
    """The protocol-identical Python server (ROCPS_PURE_PY_SERVER=1)."""
    monkeypatch.setenv("ROCPS_PURE_PY_SERVER", "1")
    ports = free_ports(3)
    cs = _mk_cluster(ports)
    master = Server(cs, "ps", 0)
    try:
        assert master._native is None  # really the python path
        c = ControlClient(cs.master_address)
        c.put("x", [1, 2, 3])
        assert c.get("x") == [1, 2, 3]
        assert c.incr("n") == 1
        c.close()
    finally:
        master.stop()
