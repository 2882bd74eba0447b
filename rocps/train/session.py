"""Trainer + MonitoredTrainingSession — capability C13 of SURVEY.md §2b.

The reference drives training with ``tf.train.MonitoredTrainingSession``
(SURVEY §3.2/§3.5): chief-only init-or-restore, non-chief
wait-for-session, hook lifecycle, periodic checkpointing, recovery.
Eager recast: a ``Trainer`` owns the model/optimizer/engine and runs
one train step per call; ``MonitoredTrainingSession`` provides the
chief/non-chief startup barrier (parameter broadcast from the chief
replaces TF's "poll ready_op"), hook wiring, auto checkpoint
save/restore, and stop handling.
"""

from __future__ import annotations

import os
from typing import Any, Callable, Dict, List, Optional

import torch

from rocps.cluster import ClusterSpec, ControlClient
from rocps.engine.comm import Comm
from rocps.engine.sync_engine import SyncEngine
from rocps.train.hooks import (CheckpointSaverHook, RunContext, RunValues,
                               SessionRunHook)

__all__ = ["Trainer", "MonitoredTrainingSession"]


class Trainer:
    """Owns model + optimizer + distributed engine + global_step."""

    def __init__(self, model: torch.nn.Module, optimizer,
                 cluster: Optional[ClusterSpec] = None,
                 job_name: str = "worker", task_index: int = 0,
                 comm: Optional[Comm] = None,
                 control: Optional[ControlClient] = None,
                 mode: str = "sync", sync_mode: Optional[str] = None,
                 bucket_cap_mb: float = 25.0,
                 grad_dtype: Optional[torch.dtype] = None,
                 device: Optional[torch.device] = None,
                 num_ps_shards: Optional[int] = None):
        self.job_name = job_name
        self.task_index = task_index
        self.cluster = cluster
        self.control = control
        self.heartbeat = None
        if cluster is not None and control is None:
            self.control = ControlClient(cluster.master_address)
        if cluster is not None:
            from rocps.cluster import HeartbeatThread
            self.heartbeat = HeartbeatThread(cluster, job_name, task_index)
        self.device = device or (
            torch.device("cuda", torch.cuda.current_device())
            if torch.cuda.is_available() else torch.device("cpu"))
        self.model = model.to(self.device)
        self.optimizer = optimizer
        # the reference's backup-worker mode (SyncReplicasOptimizer with
        # replicas_to_aggregate < total): exactly-K aggregation lives on
        # the PS shard (semisync), not on collectives
        from rocps.train.sync_replicas import SyncReplicasOptimizer
        self.replicas_to_aggregate = None
        if (mode == "sync" and isinstance(optimizer, SyncReplicasOptimizer)
                and optimizer.replicas_to_aggregate
                < optimizer.total_num_replicas
                and cluster is not None and "ps" in cluster.jobs):
            mode = "async"
            self.replicas_to_aggregate = optimizer.replicas_to_aggregate
        if comm is None:
            if mode == "async":
                # reference async mode has no worker collective group:
                # each worker talks only to the PS shards. world_size=1
                # short-circuits collectives; rank carries chief-ness.
                comm = Comm(rank=task_index, world_size=1,
                            backend="gloo", device=self.device)
            elif cluster is not None:
                comm = Comm.from_cluster(cluster, job_name, task_index,
                                         self.control, device=self.device)
            elif int(os.environ.get("WORLD_SIZE", "1")) > 1:
                comm = Comm.from_env()
                self.device = comm.device
            else:
                comm = Comm(device=self.device)
        self.comm = comm
        self.mode = mode
        # remembered so elastic recovery can rebuild the engine for a
        # shrunken world (rocps/train/recovery.py)
        self._engine_cfg = dict(sync_mode=sync_mode,
                                bucket_cap_mb=bucket_cap_mb,
                                grad_dtype=grad_dtype)
        if mode == "sync":
            self.engine = SyncEngine(self.model, comm, mode=sync_mode,
                                     bucket_cap_mb=bucket_cap_mb,
                                     grad_dtype=grad_dtype)
        elif mode == "async":
            from rocps.engine.async_engine import AsyncWorkerEngine
            self.engine = AsyncWorkerEngine(
                self.model, cluster=cluster, control=self.control,
                task_index=task_index, device=self.device,
                num_shards=num_ps_shards,
                replicas_to_aggregate=self.replicas_to_aggregate)
        else:
            raise ValueError(f"unknown mode {mode!r}")
        self.global_step = 0
        self.loss_scale = 1.0

    @property
    def is_chief(self) -> bool:
        return self.comm.rank == 0

    # ------------------------------------------------------------------
    def sync_initial_state(self) -> None:
        """Chief broadcast of params + global_step (reference §3.5 init)."""
        if self.mode == "sync":
            self.engine.broadcast_parameters(src=0)
        if self.comm.is_distributed:
            t = torch.tensor([self.global_step], dtype=torch.int64,
                             device=self.comm.device
                             if self.comm.backend == "nccl" else "cpu")
            self.comm.broadcast(t, src=0)
            if self.comm.device.type == "cuda":
                torch.cuda.synchronize()
            self.global_step = int(t.item())

    def train_step(self, loss_fn: Callable[[], torch.Tensor]) -> torch.Tensor:
        """One optimization step: fwd, bwd+comm overlapped, fused apply."""
        self.engine.zero_grad()
        if hasattr(self.optimizer, "begin_step"):
            self.optimizer.begin_step()
        loss = loss_fn()
        self.engine.step(loss, self.optimizer)
        if self.mode == "async":
            # reference: global_step counts applies cluster-wide on the PS
            self.global_step = self.engine.last_global_step
        else:
            self.global_step += 1
        return loss.detach()

    # -- checkpoint state ------------------------------------------------
    def state_tensors(self, include_optimizer: bool = True) -> Dict[str, torch.Tensor]:
        """Checkpoint state snapshot. COLLECTIVE in distributed sync
        mode (buffer averaging + rs_ag optimizer-state all-gather):
        every rank must call it at the same point — CheckpointSaverHook
        does this; don't call it from only one rank."""
        if self.mode == "async" and getattr(self.engine, "_attached", False):
            # snapshot the CURRENT shard state, not the chief's last
            # pull (one apply-round stale); also refreshes global_step
            self.engine.pull()
            self.global_step = self.engine.last_global_step
        if self.mode == "sync" and hasattr(self.engine, "sync_buffers"):
            # BN running stats are rank-local between saves; average
            # them across replicas (the reference's PS-shared moving
            # stats), so the checkpoint sees cluster-wide statistics
            self.engine.sync_buffers()
        # lazily-counted stats (BatchNormAct2d num_batches_tracked)
        # fold into their buffers before collection
        for mod in self.model.modules():
            if hasattr(mod, "_flush_nbt"):
                mod._flush_nbt()
        out: Dict[str, torch.Tensor] = {
            name: p.detach() for name, p in self.model.named_parameters()}
        for name, b in self.model.named_buffers():
            out[name] = b.detach()
        out["global_step"] = torch.tensor(self.global_step, dtype=torch.int64)
        if include_optimizer and self.mode == "sync":
            out.update(self._gather_optimizer_state())
        elif include_optimizer and self.mode == "async" and getattr(
                self.engine, "_attached", False):
            # the reference checkpoints the optimizer slot variables
            # living on the PS; fetch each shard's slot state
            for s in range(self.engine.num_shards):
                keys, tensors, t = self.engine.fetch_opt_state(s)
                for k, v in zip(keys, tensors):
                    out[f"optimizer/shard{s}/{k}"] = v
                if t:
                    out["optimizer/t"] = torch.tensor(t, dtype=torch.int64)
        return out

    def _gather_optimizer_state(self) -> Dict[str, torch.Tensor]:
        out: Dict[str, torch.Tensor] = {}
        eng: SyncEngine = self.engine  # type: ignore[assignment]
        # SyncReplicasOptimizer delegates apply_flat to .base, so the
        # slot state (momentum/adam buffers) lives on the base optimizer
        opt = getattr(self.optimizer, "base", self.optimizer)
        if hasattr(opt, "_t"):  # Adam bias-correction step counter
            out["optimizer/t"] = torch.tensor(int(opt._t),
                                              dtype=torch.int64)
        for b in eng.buckets:
            st = opt._state.get(("b", b.index))
            if not st:
                continue
            for k, v in st.items():
                if eng.mode == "rs_ag" and self.comm.is_distributed:
                    full = torch.empty(b.numel_padded, dtype=v.dtype,
                                       device=v.device)
                    self.comm.all_gather(full, v)
                    out[f"optimizer/bucket{b.index}/{k}"] = full
                else:
                    out[f"optimizer/bucket{b.index}/{k}"] = v
        return out

    def load_state_tensors(self, tensors: Dict[str, torch.Tensor]) -> None:
        params = dict(self.model.named_parameters())
        bufs = dict(self.model.named_buffers())
        for name, t in tensors.items():
            if name == "global_step":
                self.global_step = int(t.item())
            elif name in params:
                with torch.no_grad():
                    params[name].copy_(t.to(params[name].device))
            elif name in bufs:
                with torch.no_grad():
                    bufs[name].copy_(t.to(bufs[name].device))
        if self.mode == "sync":
            self._load_optimizer_state(tensors)
        elif self.mode == "async":
            # stash PS slot state; the engine pushes it after the force
            # init_shard re-seeds the shard params (chief restore path)
            restored: Dict[int, tuple] = {}
            t = int(tensors["optimizer/t"].item()) \
                if "optimizer/t" in tensors else self.global_step
            for name, v in tensors.items():
                if not name.startswith("optimizer/shard"):
                    continue
                sh, key = name[len("optimizer/shard"):].split("/", 1)
                keys, blobs, _ = restored.setdefault(int(sh), ([], [], t))
                keys.append(key)
                blobs.append(v.to(torch.float32).contiguous()
                             .numpy().tobytes())
            if restored:
                self.engine.restored_opt_state = restored

    def _load_optimizer_state(self, tensors: Dict[str, torch.Tensor]) -> None:
        eng: SyncEngine = self.engine  # type: ignore[assignment]
        base = getattr(self.optimizer, "base", self.optimizer)
        if "optimizer/t" in tensors and hasattr(base, "_t"):
            base._t = int(tensors["optimizer/t"].item())
        for b in eng.buckets:
            keys = [k for k in tensors
                    if k.startswith(f"optimizer/bucket{b.index}/")]
            if not keys:
                continue
            target = (b.my_shard(b.params_flat, self.comm.rank)
                      if eng.mode == "rs_ag" else b.params_flat)
            st = getattr(self.optimizer, "base",
                         self.optimizer).slot_state(("b", b.index), target)
            for key in keys:
                k = key.rsplit("/", 1)[1]
                if k not in st:
                    continue
                full = tensors[key].to(st[k].device)
                src = (b.my_shard(full, self.comm.rank)
                       if eng.mode == "rs_ag" and self.comm.is_distributed
                       else full)
                st[k].copy_(src.reshape(st[k].shape))


class MonitoredTrainingSession:
    """Context manager mirroring ``tf.train.MonitoredTrainingSession``."""

    def __init__(self, trainer: Trainer, master: str = "",
                 is_chief: Optional[bool] = None,
                 checkpoint_dir: Optional[str] = None,
                 hooks: Optional[List[SessionRunHook]] = None,
                 chief_only_hooks: Optional[List[SessionRunHook]] = None,
                 save_checkpoint_secs: Optional[float] = 600.0,
                 save_checkpoint_steps: Optional[int] = None,
                 max_to_keep: int = 5):
        self.trainer = trainer
        self.master = master
        self.is_chief = trainer.is_chief if is_chief is None else is_chief
        self.checkpoint_dir = checkpoint_dir
        self._stop = False
        from rocps.train.saver import Saver
        self.saver = Saver(trainer, max_to_keep=max_to_keep)
        self._hooks: List[SessionRunHook] = list(hooks or [])
        if self.is_chief:
            self._hooks += list(chief_only_hooks or [])
        # the saver hook runs on EVERY rank (state collection is a
        # collective in the reduce-scatter engine; only the chief
        # writes) — see CheckpointSaverHook
        if checkpoint_dir and not any(
                isinstance(h, CheckpointSaverHook) for h in self._hooks):
            if save_checkpoint_steps is not None:
                self._hooks.append(CheckpointSaverHook(
                    checkpoint_dir, save_steps=save_checkpoint_steps))
            elif save_checkpoint_secs is not None:
                self._hooks.append(CheckpointSaverHook(
                    checkpoint_dir, save_secs=save_checkpoint_secs))

    # -- properties hooks use -------------------------------------------
    @property
    def global_step(self) -> int:
        return self.trainer.global_step

    # -- lifecycle -------------------------------------------------------
    def __enter__(self) -> "MonitoredTrainingSession":
        from rocps.train.saver import latest_checkpoint
        if self.checkpoint_dir:
            # EVERY rank restores (not just the chief): in the reference
            # the optimizer's slot variables live on the PS, so a chief
            # restore covers all workers; here sync-mode optimizer state
            # is sharded/replicated per rank and each rank must load its
            # own piece — params/global_step are (re)broadcast below.
            ckpt = latest_checkpoint(self.checkpoint_dir)
            if ckpt:
                self.saver.restore(self.trainer, ckpt)
                if self.is_chief:
                    # async mode: the restored weights must overwrite an
                    # already-initialized PS shard (a plain chief restart
                    # without a checkpoint instead ADOPTS the PS state —
                    # reference SessionManager.prepare_session
                    # semantics); the restored global_step continues on
                    # the PS counter
                    self.trainer.engine.force_init_load = True
                    self.trainer.engine.init_global_step = \
                        self.trainer.global_step
        self.trainer.sync_initial_state()
        for h in self._hooks:
            h.begin()
        for h in self._hooks:
            h.after_create_session(self)
        return self

    def run(self, step_fn: Callable[[], Any]) -> Any:
        """One monitored step: hooks around ``trainer.train_step(step_fn)``
        (or any callable returning the step's loss)."""
        ctx = RunContext(self)
        for h in self._hooks:
            h.before_run(ctx)
        results = (self.trainer.train_step(step_fn)
                   if not callable(getattr(step_fn, "run_step", None))
                   else step_fn.run_step())
        rv = RunValues(results)
        for h in self._hooks:
            h.after_run(ctx, rv)
        if ctx._stop_requested:
            self._stop = True
        return results

    def should_stop(self) -> bool:
        return self._stop

    def request_stop(self) -> None:
        self._stop = True

    def __exit__(self, exc_type, exc, tb) -> bool:
        for h in self._hooks:
            try:
                h.end(self)
            except Exception:
                if exc is None:
                    raise
        return False

    def close(self) -> None:
        pass
