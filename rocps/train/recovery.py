"""Sync-mode in-process elastic recovery — SURVEY §5 failure handling.

The reference surfaces peer death as gRPC ``UnavailableError`` and its
``_RecoverableSession`` recreates the session, with the chief restoring
the latest checkpoint (monitored_session.py semantics).  The rocps sync
engine runs on a collective group, so the same contract becomes: when a
worker dies mid-run, the SURVIVING ranks

1. abort the broken torch.distributed group,
2. re-rendezvous through the control plane as a smaller world
   (roll-call keyed by cluster task index; lowest surviving task index
   becomes the new chief and picks a fresh TCP store port),
3. rebuild the sync engine for the new world size (reduce-scatter
   shards are world-sized),
4. restore the latest checkpoint on every rank (steps since the last
   save are lost — exactly the reference's restore-on-recreate
   contract) and re-broadcast initial state.

Failure detection is two-sided: a collective on the broken group raises
(gloo surfaces a reset/timeout, RCCL an async error), and the heartbeat
table flags the silent peer so ranks that had not yet entered the
collective do not block on it.  Chief (task 0) or control-plane death
is out of scope — that is a full restart from checkpoint, also matching
the reference (PS death → restart).
"""

from __future__ import annotations

import time
from typing import List, Optional

import torch.distributed as dist

__all__ = ["WorkerLost", "ElasticRecovery"]


class WorkerLost(RuntimeError):
    """Raised (or caught) when a peer rank died mid-step."""


class ElasticRecovery:
    def __init__(self, trainer, checkpoint_dir: Optional[str],
                 comm_timeout: float = 60.0,
                 dead_threshold: float = 6.0, join_cap: float = 45.0):
        if getattr(trainer, "mode", "sync") != "sync":
            raise ValueError(
                "ElasticRecovery covers the collective sync engine; "
                "async/semisync worker loss is tolerated by the PS shard "
                "itself (heartbeats + rejoin, rocps/engine/async_engine)")
        self.trainer = trainer
        self.checkpoint_dir = checkpoint_dir
        self.comm_timeout = comm_timeout
        self.dead_threshold = dead_threshold
        self.join_cap = join_cap
        # tasks in the CURRENT collective group; peers that died in an
        # earlier generation stay excluded (no restart-rejoin in sync
        # mode — matching the reference, a restarted worker comes back
        # through a full job restart)
        self.members = set(trainer.cluster.task_indices("worker"))

    # ------------------------------------------------------------------
    def check_peers(self) -> None:
        """Cheap pre-step probe: raise WorkerLost if a CURRENT group
        member went silent (so ranks not yet inside the collective
        don't block on it)."""
        t = self.trainer
        if t.heartbeat is None:
            return
        dead = {int(m.split(":")[1])
                for m in t.heartbeat.dead_members(self.dead_threshold)
                if m.startswith("worker:")}
        lost = (dead & self.members) - {t.task_index}
        if lost:
            raise WorkerLost(f"silent peers: {sorted(lost)}")

    # ------------------------------------------------------------------
    def _dead(self) -> set:
        t = self.trainer
        if t.heartbeat is None:
            return set()
        return {int(m.split(":")[1])
                for m in t.heartbeat.dead_members(self.dead_threshold)
                if m.startswith("worker:")}

    def _roll_call(self, epoch: int) -> List[int]:
        """Post presence under the agreed epoch, then wait until every
        current member has either posted or been declared dead (capped;
        a peer that is alive per heartbeats is WAITED FOR — its broken
        collective may take seconds to surface)."""
        t = self.trainer
        c = t.control
        c.put(f"regroup/{epoch}/task{t.task_index}", 1)
        expected = sorted(self.members)
        deadline = time.time() + self.join_cap
        alive = {t.task_index}
        while True:
            dead = self._dead()
            for i in expected:
                if i in alive:
                    continue
                try:
                    c.get(f"regroup/{epoch}/task{i}", blocking=False)
                    alive.add(i)
                except KeyError:
                    pass
            # exit when every member is accounted for: posted (alive)
            # or heartbeat-declared dead.  A live straggler keeps
            # heartbeating from its daemon thread even while its main
            # thread is stuck in a failing collective, so it is waited
            # for rather than dropped.
            if all(i in alive or i in dead for i in expected):
                break
            if time.time() > deadline:
                break
            time.sleep(0.2)
        return sorted(alive)

    # ------------------------------------------------------------------
    def recover(self) -> None:
        """Reform the group from the survivors and restore."""
        from rocps.engine.comm import Comm, _free_port
        from rocps.engine.sync_engine import SyncEngine
        from rocps.train.saver import latest_checkpoint

        t = self.trainer
        try:
            if dist.is_initialized():
                dist.destroy_process_group()
        except Exception:
            pass

        # epoch agreement through the control plane: every survivor of
        # the SAME failure reads the same counter; the leader bumps it
        # only after the assignment is published, so a straggler that
        # enters recovery late still lands in the same room
        epoch = t.control.read_counter("regroup_epoch") + 1
        self.generation = epoch
        alive = self._roll_call(epoch)
        if len(alive) < 1 or t.task_index not in alive:
            raise WorkerLost("roll call failed during regroup")
        new_rank = alive.index(t.task_index)
        world = len(alive)
        key = f"regroup/{epoch}/master"
        if new_rank == 0:
            host = t.cluster.task_address(
                "worker", t.task_index).rsplit(":", 1)[0]
            if host in ("0.0.0.0", ""):
                host = "127.0.0.1"
            addr = f"{host}:{_free_port()}"
            t.control.put(key, addr)
            t.control.put_counter("regroup_epoch", epoch)
        else:
            addr = t.control.get(key, timeout=self.comm_timeout)
        self.members = set(alive)

        backend = t.comm.backend
        device = t.comm.device
        if world > 1:
            import datetime
            Comm.tune_env()
            dist.init_process_group(
                backend=backend, init_method=f"tcp://{addr}",
                rank=new_rank, world_size=world,
                timeout=datetime.timedelta(seconds=self.comm_timeout))
        t.comm = Comm(new_rank, world, backend, device)

        # rebuild the engine: rs_ag shard sizes are world-sized, and the
        # old engine's grad hooks must come off the parameters first
        old = t.engine
        if hasattr(old, "_hooks"):
            for h in old._hooks:
                try:
                    h.remove()
                except Exception:
                    pass
        cfg = getattr(t, "_engine_cfg", {})
        t.engine = SyncEngine(t.model, t.comm,
                              mode=cfg.get("sync_mode"),
                              bucket_cap_mb=cfg.get("bucket_cap_mb", 25.0),
                              grad_dtype=cfg.get("grad_dtype"))
        # optimizer slot state is shard-shaped: drop it and reload from
        # the checkpoint (the reference restores ALL variables,
        # including slots, on recreate)
        base = getattr(t.optimizer, "base", t.optimizer)
        if hasattr(base, "_state"):
            base._state.clear()

        ckpt = (latest_checkpoint(self.checkpoint_dir)
                if self.checkpoint_dir else None)
        if ckpt:
            from rocps.train.saver import Saver
            Saver(t).restore(t, ckpt)
        t.sync_initial_state()
