"""ClusterSpec / Server bootstrap and the TCP control plane.

Capability C1 of SURVEY.md §2b: the reference describes a cluster with
``tf.train.ClusterSpec({"ps": [...], "worker": [...]})`` and starts one
``tf.train.Server`` (gRPC Master/Worker services) per process; PS
processes block in ``server.join()``.

MI355X-native recast: every process binds its cluster address and serves
a length-prefixed-JSON TCP protocol (this module). The first PS task
(or worker 0 when there is no PS job) additionally acts as the cluster
*coordination master*: key-value store (rendezvous for the RCCL/gloo
process group), named barriers, atomic counters (``global_step``),
heartbeats. Async-mode PS tasks serve parameter-shard push/pull on the
same socket (``rocps.engine.async_engine`` registers the handlers).

The data path never goes through this plane in sync mode — gradients
move via RCCL collectives over xGMI; this is bootstrap + bookkeeping.
"""

from __future__ import annotations

import json
import os
import socket
import socketserver
import struct
import threading
import time
from typing import Any, Callable, Dict, List, Optional, Union

__all__ = ["ClusterSpec", "Server", "ControlClient", "shutdown_cluster",
           "HeartbeatThread"]

_LEN = struct.Struct("<Q")
_MAX_MSG = 1 << 30


# ---------------------------------------------------------------------------
# Wire helpers: length-prefixed JSON header + optional raw binary payload.
# Message = <8B total_len><8B json_len><json bytes><payload bytes>.
# ---------------------------------------------------------------------------

def _send_msg(sock: socket.socket, obj: Dict[str, Any], payload: bytes = b"") -> None:
    head = json.dumps(obj).encode()
    sock.sendall(_LEN.pack(len(head) + len(payload) + 8) + _LEN.pack(len(head)))
    sock.sendall(head)
    if payload:
        sock.sendall(payload)


def _recv_exact(sock: socket.socket, n: int) -> bytes:
    buf = bytearray(n)
    view = memoryview(buf)
    got = 0
    while got < n:
        r = sock.recv_into(view[got:], n - got)
        if r == 0:
            raise ConnectionError("peer closed")
        got += r
    return bytes(buf)


def _recv_msg(sock: socket.socket) -> tuple[Dict[str, Any], bytes]:
    total = _LEN.unpack(_recv_exact(sock, 8))[0]
    if total > _MAX_MSG:
        raise ValueError(f"message too large: {total}")
    jlen = _LEN.unpack(_recv_exact(sock, 8))[0]
    head = json.loads(_recv_exact(sock, jlen))
    payload = _recv_exact(sock, total - 8 - jlen) if total > 8 + jlen else b""
    return head, payload


# ---------------------------------------------------------------------------
# ClusterSpec
# ---------------------------------------------------------------------------

class ClusterSpec:
    """Job-name -> task-address map; API mirrors ``tf.train.ClusterSpec``."""

    def __init__(self, cluster: Union["ClusterSpec", Dict[str, Any]]):
        if isinstance(cluster, ClusterSpec):
            self._spec: Dict[str, Dict[int, str]] = {
                j: dict(t) for j, t in cluster._spec.items()}
        elif isinstance(cluster, dict):
            self._spec = {}
            for job, tasks in cluster.items():
                if isinstance(tasks, dict):
                    self._spec[job] = {int(i): str(a) for i, a in tasks.items()}
                else:
                    self._spec[job] = {i: str(a) for i, a in enumerate(tasks)}
        else:
            raise TypeError(f"cannot build ClusterSpec from {type(cluster)}")

    @property
    def jobs(self) -> List[str]:
        return sorted(self._spec)

    def num_tasks(self, job_name: str) -> int:
        return len(self._spec[job_name])

    def task_indices(self, job_name: str) -> List[int]:
        return sorted(self._spec[job_name])

    def task_address(self, job_name: str, task_index: int) -> str:
        return self._spec[job_name][task_index]

    def job_tasks(self, job_name: str) -> List[str]:
        tasks = self._spec[job_name]
        return [tasks[i] for i in sorted(tasks)]

    def as_dict(self) -> Dict[str, List[str]]:
        return {j: self.job_tasks(j) for j in self.jobs}

    def __bool__(self) -> bool:
        return bool(self._spec)

    def __eq__(self, other: object) -> bool:
        return isinstance(other, ClusterSpec) and self._spec == other._spec

    def __repr__(self) -> str:
        return f"ClusterSpec({self.as_dict()!r})"

    # -- rocps-specific -----------------------------------------------------
    @property
    def master_job(self) -> str:
        """Job hosting the coordination master: 'ps' if present else first job."""
        return "ps" if "ps" in self._spec else self.jobs[0]

    @property
    def master_address(self) -> str:
        return self.task_address(self.master_job, self.task_indices(self.master_job)[0])

    def num_total_tasks(self) -> int:
        return sum(len(t) for t in self._spec.values())


def _split_hostport(addr: str) -> tuple[str, int]:
    host, _, port = addr.rpartition(":")
    return host or "0.0.0.0", int(port)


# ---------------------------------------------------------------------------
# Control-plane state (lives in the master process's Server)
# ---------------------------------------------------------------------------

class _MasterState:
    def __init__(self) -> None:
        self.lock = threading.Condition()
        self.kv: Dict[str, Any] = {}
        self.counters: Dict[str, int] = {}
        self.barriers: Dict[str, Dict[str, Any]] = {}
        self.members: Dict[str, float] = {}  # "job:task" -> last heartbeat
        self.shutdown = False

    # each handler returns (reply_head, reply_payload)
    def handle(self, head: Dict[str, Any], payload: bytes) -> tuple[Dict[str, Any], bytes]:
        op = head["op"]
        fn = getattr(self, "_op_" + op, None)
        if fn is None:
            return {"ok": False, "error": f"unknown op {op!r}"}, b""
        return fn(head, payload)

    def _op_register(self, head, payload):
        with self.lock:
            self.members[f"{head['job']}:{head['task']}"] = time.time()
            self.lock.notify_all()
        return {"ok": True}, b""

    def _op_heartbeat(self, head, payload):
        with self.lock:
            self.members[f"{head['job']}:{head['task']}"] = time.time()
        return {"ok": True}, b""

    def _op_members(self, head, payload):
        with self.lock:
            return {"ok": True, "members": dict(self.members)}, b""

    def _op_put(self, head, payload):
        with self.lock:
            self.kv[head["key"]] = head.get("value")
            self.lock.notify_all()
        return {"ok": True}, b""

    def _op_get(self, head, payload):
        deadline = time.time() + float(head.get("timeout", 0.0))
        with self.lock:
            while head.get("blocking") and head["key"] not in self.kv:
                remaining = deadline - time.time()
                if remaining <= 0 or self.shutdown:
                    break
                self.lock.wait(min(remaining, 1.0))
            if head["key"] in self.kv:
                return {"ok": True, "value": self.kv[head["key"]], "found": True}, b""
            return {"ok": True, "found": False}, b""

    def _op_incr(self, head, payload):
        with self.lock:
            v = self.counters.get(head["key"], 0) + int(head.get("amount", 1))
            self.counters[head["key"]] = v
        return {"ok": True, "value": v}, b""

    def _op_read_counter(self, head, payload):
        with self.lock:
            return {"ok": True, "value": self.counters.get(head["key"], 0)}, b""

    def _op_put_counter(self, head, payload):
        with self.lock:
            self.counters[head["key"]] = int(head["value"])
        return {"ok": True}, b""

    def _op_barrier(self, head, payload):
        """Named barrier: blocks until `count` distinct participants arrive."""
        name, who, count = head["name"], head["who"], int(head["count"])
        timeout = float(head.get("timeout", 300.0))
        deadline = time.time() + timeout
        with self.lock:
            b = self.barriers.setdefault(name, {"arrived": set(), "gen": 0})
            gen = b["gen"]
            b["arrived"].add(who)
            if len(b["arrived"]) >= count:
                b["gen"] += 1
                b["arrived"] = set()
                self.lock.notify_all()
                return {"ok": True}, b""
            while b["gen"] == gen and not self.shutdown:
                remaining = deadline - time.time()
                if remaining <= 0:
                    return {"ok": False, "error": f"barrier {name!r} timeout"}, b""
                self.lock.wait(min(remaining, 1.0))
            return {"ok": True}, b""

    def _op_shutdown(self, head, payload):
        with self.lock:
            self.shutdown = True
            self.lock.notify_all()
        return {"ok": True}, b""


# ---------------------------------------------------------------------------
# Server
# ---------------------------------------------------------------------------

class _Handler(socketserver.BaseRequestHandler):
    def handle(self) -> None:
        self.request.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        try:
            while True:
                head, payload = _recv_msg(self.request)
                server: "Server" = self.server.rocps_server  # type: ignore[attr-defined]
                reply, rpayload = server._dispatch(head, payload)
                _send_msg(self.request, reply, rpayload)
        except (ConnectionError, OSError):
            pass


class _TCPServer(socketserver.ThreadingTCPServer):
    allow_reuse_address = True
    daemon_threads = True


class Server:
    """One per process; mirrors ``tf.train.Server(cluster, job_name, task_index)``.

    Binds this task's cluster address and serves the control protocol.
    The master task additionally hosts cluster-wide coordination state.
    Extra services (async-PS shard push/pull) register op handlers via
    :meth:`register_handler`.

    The server itself is native (csrc/control/control_plane.cc — the C1
    [NATIVE] obligation): coordination ops run in C++ threads with no
    GIL; unknown ops call back into the registered Python handlers.
    A pure-Python server (``ROCPS_PURE_PY_SERVER=1`` or missing
    extension) keeps the same protocol for reference and tests.
    """

    def __init__(self, cluster: Union[ClusterSpec, Dict[str, Any]],
                 job_name: str, task_index: int,
                 protocol: str = "rocps", start: bool = True):
        self.cluster = ClusterSpec(cluster)
        self.job_name = job_name
        self.task_index = int(task_index)
        self.protocol = protocol
        addr = self.cluster.task_address(job_name, self.task_index)
        self._host, self._port = _split_hostport(addr)
        self.is_master = (job_name == self.cluster.master_job
                          and self.task_index == self.cluster.task_indices(job_name)[0])
        self._state = _MasterState() if self.is_master else None
        self._handlers: Dict[str, Callable] = {}
        self._shutdown_evt = threading.Event()
        self._tcp: Optional[_TCPServer] = None
        self._native = None
        self._thread: Optional[threading.Thread] = None
        if start:
            self.start()

    # -- lifecycle ----------------------------------------------------------
    def start(self) -> None:
        if self._tcp is not None or self._native is not None:
            return
        bind_host = os.environ.get("ROCPS_BIND_HOST", "0.0.0.0")
        native = None
        if not os.environ.get("ROCPS_PURE_PY_SERVER"):
            try:
                from rocps import _ccp
                native = _ccp.ControlPlaneServer(self._port, self.is_master,
                                 self.job_name, self.task_index)
            except ImportError:
                native = None
        if native is not None:
            native.set_fallback(self._native_fallback)
            self._port = native.start("" if bind_host == "0.0.0.0"
                                      else bind_host)
            self._native = native
            self._watch = threading.Thread(target=self._watch_native,
                                           daemon=True,
                                           name="rocps-native-watch")
            self._watch.start()
            # stop native threads before interpreter finalization: a C++
            # thread re-acquiring the GIL after Py_Finalize aborts
            import atexit
            atexit.register(self.stop)
        else:
            self._tcp = _TCPServer((bind_host, self._port), _Handler)
            if self._port == 0:  # ephemeral: record actual port
                self._port = self._tcp.server_address[1]
            self._tcp.rocps_server = self  # type: ignore[attr-defined]
            self._thread = threading.Thread(
                target=self._tcp.serve_forever, name="rocps-server",
                daemon=True)
            self._thread.start()
        # Register with the cluster master (self-registration included).
        # Few retries only: a master that is not up yet must not stall
        # this task's startup — heartbeats re-register within seconds.
        try:
            ControlClient(self.cluster.master_address,
                          connect_retries=8).register(
                self.job_name, self.task_index)
        except Exception:
            pass  # master may not be up yet; heartbeats will register later

    def _watch_native(self) -> None:
        self._native.wait_shutdown(0.0)
        self._shutdown_evt.set()

    def _native_fallback(self, head_bytes: bytes, payload: bytes):
        """Called from a C++ connection thread for non-builtin ops."""
        try:
            head = json.loads(head_bytes)
        except Exception:
            return (json.dumps({"ok": False, "error": "bad json"}).encode(),
                    b"")
        op = head.get("op", "")
        fn = self._handlers.get(op)
        if fn is None:
            return (json.dumps(
                {"ok": False,
                 "error": f"not master; no handler for {op!r}"}).encode(),
                b"")
        try:
            reply, rpayload = fn(head, payload)
            return json.dumps(reply).encode(), bytes(rpayload or b"")
        except Exception as e:  # surface handler errors to the caller
            return (json.dumps({"ok": False, "error": repr(e)}).encode(), b"")

    @property
    def target(self) -> str:
        return f"rocps://{self._host}:{self._port}"

    @property
    def bound_port(self) -> int:
        return self._port

    def join(self, timeout: Optional[float] = None) -> None:
        """Block until the cluster is shut down (the reference's ps loop)."""
        self._shutdown_evt.wait(timeout)

    def shutdown_requested(self) -> bool:
        return self._shutdown_evt.is_set()

    def stop(self) -> None:
        self._shutdown_evt.set()
        if self._native is not None:
            self._native.stop()
            self._native = None
            # Join the watch thread before returning: it is blocked inside
            # wait_shutdown() with the GIL released, and if the interpreter
            # starts finalizing before it re-acquires the GIL, CPython
            # pthread_exit()s it mid-pybind-frame -> std::terminate. stop()
            # runs from atexit (interpreter still alive), so joining here
            # guarantees the thread is gone before Py_Finalize.
            w = getattr(self, "_watch", None)
            if w is not None and w is not threading.current_thread():
                w.join(timeout=10.0)
        if self._tcp is not None:
            self._tcp.shutdown()
            self._tcp.server_close()
            self._tcp = None

    # -- dispatch -----------------------------------------------------------
    def register_handler(self, op: str, fn: Callable) -> None:
        """fn(head, payload) -> (reply_head, reply_payload)."""
        self._handlers[op] = fn

    def _dispatch(self, head: Dict[str, Any], payload: bytes):
        op = head.get("op", "")
        if op in self._handlers:
            return self._handlers[op](head, payload)
        if op == "shutdown":
            self._shutdown_evt.set()
        if op == "ping":
            return {"ok": True, "job": self.job_name, "task": self.task_index}, b""
        if self._state is not None:
            reply = self._state.handle(head, payload)
            if op == "shutdown":
                return reply
            return reply
        return {"ok": False, "error": f"not master; no handler for {op!r}"}, b""


# ---------------------------------------------------------------------------
# Client
# ---------------------------------------------------------------------------

class ControlClient:
    """Client for any rocps Server (connect-per-call with keepalive cache)."""

    def __init__(self, address: str, timeout: float = 300.0,
                 connect_retries: int = 120):
        self._host, self._port = _split_hostport(address)
        if self._host in ("0.0.0.0", ""):
            self._host = "127.0.0.1"
        self._timeout = timeout
        self._connect_retries = max(1, connect_retries)
        self._sock: Optional[socket.socket] = None
        self._lock = threading.Lock()

    def _connect(self) -> socket.socket:
        if self._sock is None:
            last_err: Optional[Exception] = None
            for _ in range(self._connect_retries):
                try:
                    s = socket.create_connection((self._host, self._port), timeout=5.0)
                    s.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
                    s.settimeout(self._timeout)
                    self._sock = s
                    break
                except OSError as e:  # server not up yet — bootstrap race
                    last_err = e
                    time.sleep(0.25)
            else:
                raise ConnectionError(
                    f"cannot reach rocps server {self._host}:{self._port}: {last_err}")
        return self._sock

    def call(self, op: str, payload: bytes = b"", **kw: Any) -> Dict[str, Any]:
        head = {"op": op, **kw}
        with self._lock:
            for attempt in range(2):
                try:
                    s = self._connect()
                    _send_msg(s, head, payload)
                    reply, rpayload = _recv_msg(s)
                    reply["_payload"] = rpayload
                    return reply
                except (ConnectionError, OSError):
                    self.close_nolock()
                    if attempt:
                        raise
        raise RuntimeError("unreachable")

    def call_raw(self, op: str, payload: bytes = b"", **kw: Any):
        r = self.call(op, payload, **kw)
        return r, r.pop("_payload", b"")

    # -- convenience --------------------------------------------------------
    def register(self, job: str, task: int) -> None:
        self.call("register", job=job, task=task)

    def heartbeat(self, job: str, task: int) -> None:
        self.call("heartbeat", job=job, task=task)

    def members(self) -> Dict[str, float]:
        return self.call("members")["members"]

    def put(self, key: str, value: Any) -> None:
        self.call("put", key=key, value=value)

    def get(self, key: str, blocking: bool = True, timeout: float = 120.0) -> Any:
        r = self.call("get", key=key, blocking=blocking, timeout=timeout)
        if not r.get("found"):
            raise KeyError(key)
        return r["value"]

    def incr(self, key: str, amount: int = 1) -> int:
        return self.call("incr", key=key, amount=amount)["value"]

    def read_counter(self, key: str) -> int:
        return self.call("read_counter", key=key)["value"]

    def put_counter(self, key: str, value: int) -> None:
        self.call("put_counter", key=key, value=value)

    def barrier(self, name: str, who: str, count: int, timeout: float = 300.0) -> None:
        r = self.call("barrier", name=name, who=who, count=count, timeout=timeout)
        if not r.get("ok"):
            raise TimeoutError(r.get("error", "barrier failed"))

    def shutdown_cluster(self) -> None:
        try:
            self.call("shutdown")
        except (ConnectionError, OSError):
            pass

    def close_nolock(self) -> None:
        if self._sock is not None:
            try:
                self._sock.close()
            except OSError:
                pass
            self._sock = None

    def close(self) -> None:
        with self._lock:
            self.close_nolock()

    def __enter__(self) -> "ControlClient":
        return self

    def __exit__(self, *a: Any) -> None:
        self.close()


def shutdown_cluster(cluster: ClusterSpec) -> None:
    """Send the shutdown op to every task's server (releases join())."""
    for job in cluster.jobs:
        for i in cluster.task_indices(job):
            try:
                c = ControlClient(cluster.task_address(job, i), timeout=5.0,
                                  connect_retries=1)
                c.call("shutdown")
                c.close()
            except Exception:
                pass  # task already gone — that's what shutdown wants


class HeartbeatThread:
    """Failure detection: each task heartbeats the master;
    ``dead_members`` reports tasks silent for > threshold seconds.

    The reference surfaces peer death as gRPC UnavailableError; here the
    control plane's membership table plays that role — the async PS
    tolerates silent workers, sync-mode scripts can poll
    ``dead_members`` and trigger restart-from-checkpoint.
    """

    def __init__(self, cluster: ClusterSpec, job: str, task: int,
                 interval: float = 2.0):
        self.client = ControlClient(cluster.master_address)
        self.job, self.task = job, task
        self.interval = interval
        self._stop = threading.Event()
        self._thread = threading.Thread(target=self._run, daemon=True,
                                        name="rocps-heartbeat")
        self._thread.start()

    def _run(self) -> None:
        while not self._stop.wait(self.interval):
            try:
                self.client.heartbeat(self.job, self.task)
            except Exception:
                pass  # master unreachable — the master is the failed one

    def stop(self) -> None:
        self._stop.set()

    def dead_members(self, threshold: float = 10.0) -> List[str]:
        now = time.time()
        try:
            members = self.client.members()
        except Exception:
            return []
        return [m for m, last in members.items() if now - last > threshold]
