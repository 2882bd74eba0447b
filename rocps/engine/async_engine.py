"""Async-SGD parameter-server engine — capability C5 of SURVEY.md §2b.

Reference semantics (§3.2): each worker independently pushes gradients;
the optimizer kernels run *on the PS device* immediately (no
coordination, ``use_locking=False`` — parameter reads may be torn /
stale by design); the worker then pulls fresh parameters. A worker's
train op completes only when its own apply finished (the gRPC RunGraph
round trip), but workers never wait for each other.

MI355X-native recast (single node, 8 GPU workers + host-DRAM PS):
- Each PS task owns a *shard* of the flat parameter space in a
  POSIX shared-memory segment (/dev/shm), with per-worker gradient
  slots and seq/ack words.
- Workers mmap the segment: gradient push is one device→host copy per
  shard (hipMemcpyAsync once the region is hipHostRegister'ed; plain
  copy on CPU), then a seq bump; parameter pull is one host→device
  copy. No per-tensor RPC, no serialization — the copy engine moves
  the bytes over PCIe/IF while compute streams stay busy.
- The PS apply thread spins on seq!=ack, applies the (fused CPU)
  optimizer per worker push, bumps version + global_step (one bump
  per APPLY, the reference's AssignAdd semantics). Staleness
  accounting: each worker records the shard version it observed at
  pull time (pver); a push is dropped when the shard has advanced
  more than ``staleness_bound`` applies since that pull (the
  SyncReplicasOptimizer stale-drop rule recast per-apply, C4/C5).
  Dropped pushes are still acked so the worker re-pulls and rejoins.

Wire bootstrap (control plane): the chief publishes a manifest and
init params; each PS task serves ``init_shard`` and publishes its shm
path; workers attach.
"""

from __future__ import annotations

import mmap
import os
import threading
import time
from typing import Dict, List, Optional

import numpy as np
import torch

from rocps.cluster import ClusterSpec, ControlClient, Server
from rocps.placement import plan_shards
from rocps.train.coordinator import Coordinator

__all__ = ["AsyncPSShard", "AsyncWorkerEngine"]

_MAGIC = 0x52C0_5A51
_ALIGN = 256


def _np_dtype(t: torch.dtype):
    return {torch.float32: np.float32, torch.float64: np.float64,
            torch.bfloat16: np.uint16, torch.float16: np.float16}[t]


class _ShmLayout:
    """header int64[8 + 3*W] | params | grads[W]  (offsets 256-aligned).

    header: [magic, version(applies), num_workers, numel, dtype_code,
             global_step, round, reserved, then per worker:
             seq, ack, pver(param round the worker last pulled)]
    """

    def __init__(self, numel: int, num_workers: int, itemsize: int):
        self.numel = numel
        self.num_workers = num_workers
        self.header_words = 8 + 3 * num_workers
        self.params_off = ((self.header_words * 8 + _ALIGN - 1)
                           // _ALIGN * _ALIGN)
        grad_bytes = numel * itemsize
        self.grad_off0 = ((self.params_off + grad_bytes + _ALIGN - 1)
                          // _ALIGN * _ALIGN)
        self.grad_stride = (grad_bytes + _ALIGN - 1) // _ALIGN * _ALIGN
        self.total = self.grad_off0 + self.grad_stride * num_workers

    def grad_off(self, w: int) -> int:
        return self.grad_off0 + w * self.grad_stride

    @staticmethod
    def seq_idx(w: int) -> int:
        return 8 + 3 * w

    @staticmethod
    def ack_idx(w: int) -> int:
        return 8 + 3 * w + 1

    @staticmethod
    def pver_idx(w: int) -> int:
        return 8 + 3 * w + 2


class _ShmSeg:
    """A mapped shared-memory segment with torch views."""

    def __init__(self, path: str, layout: _ShmLayout, dtype: torch.dtype,
                 create: bool):
        self.path = path
        self.layout = layout
        self.dtype = dtype
        flags = os.O_RDWR | (os.O_CREAT if create else 0)
        self.fd = os.open(path, flags, 0o600)
        if create:
            os.ftruncate(self.fd, layout.total)
        self.mm = mmap.mmap(self.fd, layout.total)
        self.buf = np.frombuffer(self.mm, dtype=np.uint8)
        self.header = torch.from_numpy(
            np.frombuffer(self.mm, dtype=np.int64,
                          count=layout.header_words, offset=0))
        npdt = _np_dtype(dtype)
        params_np = np.frombuffer(self.mm, dtype=npdt, count=layout.numel,
                                  offset=layout.params_off)
        self.params = torch.from_numpy(params_np)
        if dtype == torch.bfloat16:
            self.params = self.params.view(torch.bfloat16)
        self.grads: List[torch.Tensor] = []
        for w in range(layout.num_workers):
            g_np = np.frombuffer(self.mm, dtype=npdt, count=layout.numel,
                                 offset=layout.grad_off(w))
            g = torch.from_numpy(g_np)
            if dtype == torch.bfloat16:
                g = g.view(torch.bfloat16)
            self.grads.append(g)

    def close(self) -> None:
        del self.header, self.params, self.grads, self.buf
        self.mm.close()
        os.close(self.fd)


# ---------------------------------------------------------------------------
# PS side
# ---------------------------------------------------------------------------

class AsyncPSShard:
    """Runs inside a ps task process: hosts one shard + its apply thread."""

    def __init__(self, server: Server, cluster: ClusterSpec,
                 shard_index: int):
        self.server = server
        self.cluster = cluster
        self.shard_index = shard_index
        self.master = ControlClient(cluster.master_address)
        self.num_workers = cluster.num_tasks("worker")
        self.coord = Coordinator()
        self.seg: Optional[_ShmSeg] = None
        self.optimizer = None
        self._apply_thread: Optional[threading.Thread] = None
        server.register_handler("init_shard", self._on_init_shard)
        server.register_handler("push_pull", self._on_push_pull)
        server.register_handler("shard_stats", self._on_stats)
        server.register_handler("opt_state_get", self._on_opt_state_get)
        server.register_handler("opt_state_put", self._on_opt_state_put)
        self.applied = 0
        self.dropped_stale = 0
        self.total_pushes = 0
        # serializes optimizer applies between the shm apply loop and
        # the TCP push_pull handler threads (remote workers)
        self._apply_mu = threading.Lock()
        self.staleness_bound: Optional[int] = None
        self.replicas_to_aggregate: Optional[int] = None
        self.flush_timeout = 2.0
        self.partial_rounds = 0

    @staticmethod
    def serve_on(server: Server, cluster: ClusterSpec) -> "AsyncPSShard":
        idx = cluster.task_indices("ps").index(server.task_index)
        return AsyncPSShard(server, cluster, idx)

    # -- op handlers ----------------------------------------------------
    def _on_init_shard(self, head: Dict, payload: bytes):
        numel = int(head["numel"])
        dtype = getattr(torch, head["dtype"])
        itemsize = torch.empty(0, dtype=dtype).element_size()
        if self.seg is not None:
            # Re-init (chief restart): the apply thread still maps the
            # segment, so never close/replace it. Reference semantics
            # (SessionManager.prepare_session): an already-initialized
            # model is ADOPTED by a restarting chief — its fresh random
            # weights must not clobber PS progress — unless the chief
            # explicitly restored a checkpoint (``force``), which wins.
            if (self.seg.layout.numel != numel
                    or self.seg.dtype != dtype):
                return ({"ok": False,
                         "error": "shard already initialized with a "
                                  "different shape"}, b"")
            if head.get("force"):
                src = np.frombuffer(payload, dtype=_np_dtype(dtype),
                                    count=numel)
                t = torch.from_numpy(src.copy())
                if dtype == torch.bfloat16:
                    t = t.view(torch.bfloat16)
                self.seg.params.copy_(t)
                self.seg.header[5] = int(head.get("global_step") or 0)
            self.master.put(f"async_shm_{self.shard_index}",
                            {"path": self.seg.path, "numel": numel,
                             "dtype": head["dtype"]})
            return {"ok": True, "path": self.seg.path,
                    "adopted": not head.get("force")}, b""
        layout = _ShmLayout(numel, self.num_workers, itemsize)
        path = f"/dev/shm/rocps_shard{self.shard_index}_{os.getpid()}"
        self.seg = _ShmSeg(path, layout, dtype, create=True)
        h = self.seg.header
        h.zero_()
        h[0] = _MAGIC
        h[2] = self.num_workers
        h[3] = numel
        # restored run: the reference's global_step lives on the PS and
        # is restored with the variables — continue counting from it
        h[5] = int(head.get("global_step") or 0)
        # initial params from chief
        src = np.frombuffer(payload, dtype=_np_dtype(dtype), count=numel)
        t = torch.from_numpy(src.copy())
        if dtype == torch.bfloat16:
            t = t.view(torch.bfloat16)
        self.seg.params.copy_(t)
        # optimizer from manifest
        opt_spec = head.get("optimizer", {"type": "sgd", "lr": 0.01})
        self.optimizer = _build_optimizer(opt_spec)
        if hasattr(self.optimizer, "_t"):
            # resumed run: Adam's bias-correction counter continues from
            # the restored step, like the reference's beta-power slots
            self.optimizer._t = int(head.get("global_step") or 0)
        self.staleness_bound = head.get("staleness_bound")
        self.replicas_to_aggregate = head.get("replicas_to_aggregate")
        self.flush_timeout = float(head.get("flush_timeout") or 2.0)
        if self._apply_thread is None:
            self._apply_thread = threading.Thread(
                target=self._apply_loop, daemon=True,
                name=f"rocps-ps-apply-{self.shard_index}")
            self._apply_thread.start()
            import atexit
            atexit.register(self._cleanup)
        self.master.put(f"async_shm_{self.shard_index}",
                        {"path": path, "numel": numel,
                         "dtype": head["dtype"]})
        return {"ok": True, "path": path}, b""

    def _cleanup(self) -> None:
        """Shutdown path (atexit): stop + JOIN the apply thread first —
        a daemon thread killed inside a torch op during interpreter
        finalization aborts the process — then unlink the /dev/shm
        segment (workers hold their own mappings)."""
        self.coord.request_stop()
        t = self._apply_thread
        if t is not None and t.is_alive():
            t.join(timeout=10.0)
        seg = self.seg
        if seg is not None:
            try:
                os.unlink(seg.path)
            except OSError:
                pass

    def _on_opt_state_get(self, head, payload):
        """Chief checkpoint path: the reference saves the optimizer's
        slot variables living on the PS; here that is the shard-slot
        state (momentum / adam m,v / fp32 master), returned as fp32
        blobs. Reads race the apply thread by design (same staleness
        as the chief's parameter pull)."""
        if self.optimizer is None:
            return {"ok": False, "error": "shard not initialized"}, b""
        st = self.optimizer._state.get("shard") or {}
        keys = sorted(k for k, v in st.items() if torch.is_tensor(v))
        blobs = [st[k].detach().to(torch.float32).contiguous()
                 .numpy().tobytes() for k in keys]
        return ({"ok": True, "keys": keys, "lens": [len(b) for b in blobs],
                 "t": int(getattr(self.optimizer, "_t", 0))},
                b"".join(blobs))

    def _on_opt_state_put(self, head, payload):
        """Checkpoint-restore path: chief pushes the saved slot state
        back (after a force init_shard re-seeded the params)."""
        if self.optimizer is None or self.seg is None:
            return {"ok": False, "error": "shard not initialized"}, b""
        st = self.optimizer.slot_state("shard", self.seg.params)
        off = 0
        for k, ln in zip(head.get("keys", []), head.get("lens", [])):
            arr = np.frombuffer(payload, dtype=np.float32, count=ln // 4,
                                offset=off)
            if k in st and torch.is_tensor(st[k]):
                st[k].copy_(torch.from_numpy(arr.copy())
                            .reshape(st[k].shape))
            off += ln
        if hasattr(self.optimizer, "_t") and head.get("t") is not None:
            self.optimizer._t = int(head["t"])
        return {"ok": True}, b""

    def _on_stats(self, head, payload):
        return {"ok": True, "applied": self.applied,
                "dropped_stale": self.dropped_stale,
                "partial_rounds": self.partial_rounds,
                "total_pushes": self.total_pushes,
                "version": int(self.seg.header[1]) if self.seg else 0}, b""

    def _on_push_pull(self, head, payload):
        """Cross-host worker data path (the reference's PS
        worked over TCP between hosts).  One round trip = gradient push
        + immediate apply + fresh-parameter pull, i.e. the reference's
        RunGraph exchange.  Remote workers use it when the shard's
        /dev/shm segment is not reachable (different host); local
        workers keep the shm fast path.  Async mode only — semisync
        round membership needs the shm seq/ack slots."""
        if self.seg is None:
            return {"ok": False, "error": "shard uninitialized"}, b""
        if self.replicas_to_aggregate:
            return {"ok": False,
                    "error": "tcp transport is async-mode only"}, b""
        seg = self.seg
        h = seg.header
        with self._apply_mu:
            if payload and not head.get("pull_only"):
                version = int(h[1])
                if (self.staleness_bound is not None
                        and version - int(head.get("pver", 0))
                        > self.staleness_bound):
                    self.dropped_stale += 1
                else:
                    src = np.frombuffer(payload,
                                        dtype=_np_dtype(seg.dtype),
                                        count=seg.layout.numel)
                    g = torch.from_numpy(src.copy())
                    if seg.dtype == torch.bfloat16:
                        g = g.view(torch.bfloat16)
                    if hasattr(self.optimizer, "begin_step"):
                        self.optimizer.begin_step()
                    self.optimizer.apply_flat(seg.params, g, slot="shard")
                    self.applied += 1
                    h[1] = version + 1
                    h[5] += 1
                self.total_pushes += 1
            view = (seg.params.view(torch.uint16)
                    if seg.dtype == torch.bfloat16 else seg.params)
            out = view.numpy().tobytes()
            return {"ok": True, "version": int(h[1]),
                    "global_step": int(h[5])}, out

    # -- the apply loop --------------------------------------------------
    def _apply_loop(self) -> None:
        if self.replicas_to_aggregate:
            self._semisync_loop()
        else:
            self._async_loop()

    def _async_loop(self) -> None:
        """Reference async mode: every push applied immediately."""
        seg = self.seg
        h = seg.header
        W = self.num_workers
        L = _ShmLayout
        with self.coord.stop_on_exception():
            while not (self.coord.should_stop()
                       or self.server.shutdown_requested()):
                progressed = False
                for w in range(W):
                    seq, ack = int(h[L.seq_idx(w)]), int(h[L.ack_idx(w)])
                    if seq != ack:
                        # stale-drop rule: drop when the shard advanced
                        # more than staleness_bound APPLIES since this
                        # worker's last pull (pver = shard version the
                        # worker observed at pull time). The dropped
                        # push is still acked, so the worker re-pulls
                        # (refreshing pver) and its next push is fresh.
                        with self._apply_mu:
                            version = int(h[1])
                            if (self.staleness_bound is not None
                                    and version - int(h[L.pver_idx(w)])
                                    > self.staleness_bound):
                                self.dropped_stale += 1
                            else:
                                if hasattr(self.optimizer, "begin_step"):
                                    self.optimizer.begin_step()
                                self.optimizer.apply_flat(
                                    seg.params, seg.grads[w], slot="shard")
                                self.applied += 1
                                h[1] = version + 1
                                # global_step advances once per APPLY
                                # (the reference's AssignAdd)
                                h[5] += 1
                            self.total_pushes += 1
                        h[L.ack_idx(w)] = seq  # ack
                        progressed = True
                if not progressed:
                    time.sleep(0.0002)

    def _semisync_loop(self) -> None:
        """SyncReplicasOptimizer semantics (C4, replicas_to_aggregate=K
        <= total): accumulate exactly K FRESH gradients per round
        (fresh = computed against this round's parameters, i.e. the
        worker's pver equals the current round — the reference's
        local_step >= global_step check), average, apply ONCE, bump the
        round, and only then ack the contributors (the token-queue
        release). Stale pushes are dropped but acked so the straggler
        re-pulls and rejoins."""
        seg = self.seg
        h = seg.header
        W = self.num_workers
        K = self.replicas_to_aggregate
        L = _ShmLayout
        accum = torch.zeros(seg.params.numel(), dtype=torch.float32)
        pending: list = []  # contributors awaiting the round apply
        n_contrib = 0
        last_contrib_t = time.time()

        def apply_round(n):
            nonlocal n_contrib, pending, last_contrib_t
            accum.div_(n)
            if hasattr(self.optimizer, "begin_step"):
                self.optimizer.begin_step()
            self.optimizer.apply_flat(seg.params, accum.to(seg.params.dtype),
                                      slot="shard")
            self.applied += 1
            h[1] += 1
            h[6] = int(h[6]) + 1  # new round
            # one aggregated apply = one global_step, matching the
            # reference SyncReplicasOptimizer (step-keyed hooks and LR
            # schedules advance per round, not per contribution)
            h[5] += 1
            accum.zero_()
            n_contrib = 0
            for pw in pending:  # token-queue release
                h[L.ack_idx(pw)] = int(h[L.seq_idx(pw)])
            pending = []
            last_contrib_t = time.time()

        with self.coord.stop_on_exception():
            while not (self.coord.should_stop()
                       or self.server.shutdown_requested()):
                progressed = False
                rnd = int(h[6])
                for w in range(W):
                    if w in pending:  # already contributed; awaiting apply
                        continue
                    seq, ack = int(h[L.seq_idx(w)]), int(h[L.ack_idx(w)])
                    if seq == ack:
                        continue
                    progressed = True
                    self.total_pushes += 1
                    if int(h[L.pver_idx(w)]) != rnd:
                        self.dropped_stale += 1   # stale round
                        h[L.ack_idx(w)] = seq     # release the straggler
                        continue
                    accum += seg.grads[w].float()
                    pending.append(w)
                    n_contrib += 1
                    last_contrib_t = time.time()
                    if n_contrib >= K:
                        apply_round(n_contrib)
                        break  # re-read the new round
                if not progressed:
                    # liveness: at end-of-training the final round may
                    # never reach K contributors — flush it so pending
                    # workers are released (counted in partial_rounds)
                    if (pending
                            and time.time() - last_contrib_t
                            > self.flush_timeout):
                        self.partial_rounds += 1
                        apply_round(n_contrib)
                        continue
                    time.sleep(0.0002)


def _build_optimizer(spec: Dict):
    from rocps.train.optimizer import (AdamOptimizer,
                                       GradientDescentOptimizer,
                                       MomentumOptimizer)
    t = spec.get("type", "sgd")
    if t == "sgd":
        return GradientDescentOptimizer(spec.get("lr", 0.01))
    if t == "momentum":
        return MomentumOptimizer(spec.get("lr", 0.01),
                                 spec.get("momentum", 0.9),
                                 spec.get("nesterov", False))
    if t == "adam":
        o = AdamOptimizer(spec.get("lr", 1e-3), spec.get("beta1", 0.9),
                          spec.get("beta2", 0.999), spec.get("eps", 1e-8))
        return o
    raise ValueError(f"unknown optimizer type {t!r}")


def optimizer_spec(opt) -> Dict:
    from rocps.train.optimizer import (AdamOptimizer,
                                       GradientDescentOptimizer,
                                       MomentumOptimizer)
    base = getattr(opt, "base", opt)
    if isinstance(base, MomentumOptimizer):
        return {"type": "momentum", "lr": base.lr, "momentum": base.momentum,
                "nesterov": base.use_nesterov}
    if isinstance(base, AdamOptimizer):
        return {"type": "adam", "lr": base.lr, "beta1": base.beta1,
                "beta2": base.beta2, "eps": base.epsilon}
    if isinstance(base, GradientDescentOptimizer):
        return {"type": "sgd", "lr": base.lr}
    return {"type": "sgd", "lr": getattr(base, "lr", 0.01)}


# ---------------------------------------------------------------------------
# Worker side
# ---------------------------------------------------------------------------

class AsyncWorkerEngine:
    """Worker-side engine: flat per-shard grad buffers; push/ack/pull."""

    def __init__(self, model: torch.nn.Module, cluster: ClusterSpec,
                 control: ControlClient, task_index: int,
                 device: torch.device, num_shards: Optional[int] = None,
                 strategy: str = "greedy",
                 staleness_bound: Optional[int] = None,
                 replicas_to_aggregate: Optional[int] = None,
                 flush_timeout: float = 2.0):
        self.model = model
        self.cluster = cluster
        self.control = control
        self.task_index = task_index
        self.device = device
        self.num_shards = num_shards or cluster.num_tasks("ps")
        self.staleness_bound = staleness_bound
        self.replicas_to_aggregate = replicas_to_aggregate
        self.flush_timeout = flush_timeout
        named = [(n, p) for n, p in model.named_parameters()
                 if p.requires_grad]
        self.plan = plan_shards(named, self.num_shards, strategy=strategy)
        # group params by shard, keep deterministic order
        self.shard_params: List[List] = [[] for _ in range(self.num_shards)]
        for n, p in named:
            self.shard_params[self.plan.shard_of(n)].append((n, p))
        self.dtype = named[0][1].dtype
        # local flat param/grad buffers per shard; param/grad views wired
        self.flat_params: List[torch.Tensor] = []
        self.flat_grads: List[torch.Tensor] = []
        for s in range(self.num_shards):
            numel = sum(p.numel() for _, p in self.shard_params[s])
            fp = torch.zeros(numel, dtype=self.dtype, device=device)
            fg = torch.zeros(numel, dtype=self.dtype, device=device)
            off = 0
            for n, p in self.shard_params[s]:
                k = p.numel()
                fp[off:off + k] = p.data.reshape(-1)
                p.data = fp[off:off + k].view_as(p.data)
                p.grad = fg[off:off + k].view_as(p.data)
                off += k
            self.flat_params.append(fp)
            self.flat_grads.append(fg)
        self.segs: List[Optional[_ShmSeg]] = [None] * self.num_shards
        # cross-host shards: ControlClient data path instead of shm
        self._tcp: Dict[int, ControlClient] = {}
        self._tcp_meta: Dict[int, tuple] = {}
        self._tcp_pver: Dict[int, int] = {}
        # side stream: push/pull copies run off the compute stream (C7's
        # third stream) so pinned DMA overlaps across shards
        self._side = (torch.cuda.Stream(device=device)
                      if device.type == "cuda" else None)
        self.local_step = 0
        self.is_chief = task_index == 0
        self._attached = False
        self.last_global_step = 0

    # -- bootstrap -------------------------------------------------------
    def initialize(self, optimizer, force_load: Optional[bool] = None) -> None:
        """Chief: init_shard every PS shard. ``force_load`` (or the
        ``force_init_load`` attribute, set by the session after a
        checkpoint restore) makes the pushed params overwrite an
        already-initialized shard; otherwise a re-init adopts the PS
        state (reference SessionManager.prepare_session semantics)."""
        force = (force_load if force_load is not None
                 else getattr(self, "force_init_load", False))
        if self.is_chief:
            for s in range(self.num_shards):
                ps_addr = self.cluster.task_address(
                    "ps", self.cluster.task_indices("ps")[s])
                c = ControlClient(ps_addr)
                flat = self.flat_params[s].detach().cpu()
                view = (flat.view(torch.uint16) if self.dtype == torch.bfloat16
                        else flat)
                payload = view.numpy().tobytes()
                deadline = time.time() + 60.0
                while True:
                    r = c.call("init_shard", payload=payload,
                               numel=flat.numel(),
                               dtype=str(self.dtype).replace("torch.", ""),
                               optimizer=optimizer_spec(optimizer),
                               staleness_bound=self.staleness_bound,
                               replicas_to_aggregate=self.replicas_to_aggregate,
                               flush_timeout=self.flush_timeout,
                               force=bool(force),
                               global_step=int(getattr(
                                   self, "init_global_step", 0)))
                    if r.get("ok"):
                        break
                    # PS may still be wiring its shard service (startup race)
                    if ("no handler" in str(r.get("error", ""))
                            and time.time() < deadline):
                        time.sleep(0.2)
                        continue
                    raise RuntimeError(f"init_shard failed: {r}")
                restored = getattr(self, "restored_opt_state", None)
                if force and restored and s in restored:
                    keys, blobs, t = restored[s]
                    r = c.call("opt_state_put", payload=b"".join(blobs),
                               keys=keys, lens=[len(b) for b in blobs],
                               t=t)
                    if not r.get("ok"):
                        raise RuntimeError(f"opt_state_put failed: {r}")
                c.close()
        self._attach()

    def fetch_opt_state(self, shard: int):
        """(keys, fp32 tensors, t) of the PS shard's optimizer slots."""
        ps_addr = self.cluster.task_address(
            "ps", self.cluster.task_indices("ps")[shard])
        c = ControlClient(ps_addr)
        try:
            r, payload = c.call_raw("opt_state_get")
        finally:
            c.close()
        if not r.get("ok"):
            return [], [], 0
        out, off = [], 0
        for ln in r["lens"]:
            arr = np.frombuffer(payload, dtype=np.float32, count=ln // 4,
                                offset=off)
            out.append(torch.from_numpy(arr.copy()))
            off += ln
        return list(r["keys"]), out, int(r.get("t", 0))

    def _attach(self) -> None:
        force_tcp = os.environ.get("ROCPS_ASYNC_TCP") == "1"
        for s in range(self.num_shards):
            info = self.control.get(f"async_shm_{s}", blocking=True,
                                    timeout=120.0)
            dtype = getattr(torch, info["dtype"])
            itemsize = torch.empty(0, dtype=dtype).element_size()
            layout = _ShmLayout(int(info["numel"]),
                                self.cluster.num_tasks("worker"), itemsize)
            seg = None
            if not force_tcp:
                try:
                    seg = _ShmSeg(info["path"], layout, dtype, create=False)
                except OSError:
                    seg = None  # shard on another host: TCP data path
            if seg is None:
                # cross-host transport (the reference's PS
                # spoke TCP between hosts): push+apply+pull is one
                # push_pull round trip to the shard's control server
                ps_addr = self.cluster.task_address(
                    "ps", self.cluster.task_indices("ps")[s])
                self._tcp[s] = ControlClient(ps_addr)
                self._tcp_meta[s] = (int(info["numel"]), dtype)
                self._tcp_pver[s] = 0
                if self.device.type == "cuda":
                    pass  # params land via flat_params copy below
            else:
                if self.device.type == "cuda":
                    self._pin_segment(seg, layout)
            self.segs[s] = seg
        self._attached = True
        self.pull()

    @staticmethod
    def _pin_segment(seg: _ShmSeg, layout: _ShmLayout) -> None:
        """hipHostRegister the mapped shard so push/pull run as true
        pinned DMA on the copy engine (hipMemcpyAsync on a
        side stream). Falls back to pageable copies if registration is
        refused."""
        try:
            import ctypes
            base = ctypes.addressof(ctypes.c_char.from_buffer(seg.mm))
            r = torch.cuda.cudart().cudaHostRegister(base, layout.total, 0)
            if int(r) != 0:
                raise RuntimeError(f"cudaHostRegister rc={int(r)}")
            seg.pinned = True
        except Exception:
            seg.pinned = False

    # -- engine interface (used by Trainer) ------------------------------
    def broadcast_parameters(self, src: int = 0) -> None:
        pass  # params come from the PS pull

    def zero_grad(self) -> None:
        for g in self.flat_grads:
            g.zero_()

    def step(self, loss: torch.Tensor, optimizer) -> None:
        if not self._attached:
            self.initialize(optimizer)
        loss.backward()
        self.push_and_pull()

    def _tcp_round_trip(self, s: int, push: bool) -> None:
        numel, dtype = self._tcp_meta[s]
        payload = b""
        if push:
            flat = self.flat_grads[s].detach().to("cpu")
            view = (flat.view(torch.uint16) if dtype == torch.bfloat16
                    else flat)
            payload = view.numpy().tobytes()
        r, params = self._tcp[s].call_raw(
            "push_pull", payload=payload, worker=self.task_index,
            pver=self._tcp_pver[s], pull_only=not push)
        if not r.get("ok"):
            raise RuntimeError(f"push_pull failed on shard {s}: {r}")
        arr = np.frombuffer(params, dtype=_np_dtype(dtype), count=numel)
        t = torch.from_numpy(arr.copy())
        if dtype == torch.bfloat16:
            t = t.view(torch.bfloat16)
        self.flat_params[s].copy_(t.to(self.flat_params[s].device))
        self._tcp_pver[s] = int(r["version"])
        self.last_global_step = int(r["global_step"])

    def push_and_pull(self) -> None:
        self.local_step += 1
        w = self.task_index
        if self._tcp:
            for s in list(self._tcp):
                self._tcp_round_trip(s, push=True)
            if len(self._tcp) == self.num_shards:
                return
            # mixed shm/tcp: fall through for the shm shards only
        # push grads to every shard, bump seq (D2H straight into the
        # shard slot — pinned DMA on the side stream when _pin_segment
        # succeeded)
        if self._side is not None:
            self._side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(self._side):
                for s, seg in enumerate(self.segs):
                    if seg is None:
                        continue
                    seg.grads[w].copy_(self.flat_grads[s],
                                       non_blocking=getattr(seg, "pinned",
                                                            False))
            self._side.synchronize()  # grads visible before the seq bump
            for s, seg in enumerate(self.segs):
                if seg is not None:
                    seg.header[_ShmLayout.seq_idx(w)] = self.local_step
        else:
            for s, seg in enumerate(self.segs):
                if seg is None:
                    continue
                seg.grads[w].copy_(self.flat_grads[s])
                seg.header[_ShmLayout.seq_idx(w)] = self.local_step
        # wait for own applies (the reference's RunGraph round trip; in
        # semi-sync mode this is the token-queue wait for the round)
        deadline = time.time() + 120.0
        for seg in self.segs:
            if seg is None:
                continue
            while int(seg.header[_ShmLayout.ack_idx(w)]) != self.local_step:
                if time.time() > deadline:
                    raise TimeoutError("PS apply timeout (shard dead?)")
                time.sleep(0.0002)
        # pull fresh (possibly concurrently-updated) params
        self.pull()
        for seg in self.segs:
            if seg is not None:
                self.last_global_step = int(seg.header[5])
                break

    def pull(self) -> None:
        w = self.task_index
        # pver records the shard VERSION (header[1], count of applies)
        # observed at pull time. Async staleness compares the current
        # version against it; in semisync the version and the round
        # counter (header[6]) advance in lockstep (both bumped once per
        # apply_round), so the freshness check pver == round holds too.
        rounds = [int(seg.header[1]) if seg is not None else 0
                  for seg in self.segs]
        for s in self._tcp:
            self._tcp_round_trip(s, push=False)
        if self._side is not None:
            with torch.cuda.stream(self._side):
                for s, seg in enumerate(self.segs):
                    if seg is None:
                        continue
                    self.flat_params[s].copy_(seg.params,
                                              non_blocking=getattr(
                                                  seg, "pinned", False))
            torch.cuda.current_stream().wait_stream(self._side)
        else:
            for s, seg in enumerate(self.segs):
                if seg is not None:
                    self.flat_params[s].copy_(seg.params)
        # claim the round read BEFORE the copy: if an apply landed
        # mid-pull our grads will be (conservatively) counted stale
        for s, seg in enumerate(self.segs):
            if seg is not None:
                seg.header[_ShmLayout.pver_idx(w)] = rounds[s]

    def detach(self) -> None:
        for seg in self.segs:
            if seg is not None:
                seg.close()
