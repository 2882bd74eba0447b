"""In-process loopback collective backend — SURVEY §4.4.

N "ranks" run as threads in one process; collectives rendezvous on a
barrier and rank 0 computes the reduction once (bit-identical results
on every rank, deterministic order). Lets engine logic — bucketing,
reduce-scatter shard math, fused-optimizer-on-shard, all-gather,
staleness handling — be tested without RCCL or gloo, and supports
fault injection (a dead rank breaks the barrier like a dropped peer).
"""

from __future__ import annotations

import threading
from typing import Callable, Dict, List, Optional

import torch

from rocps.engine.comm import Comm, _Done

__all__ = ["FakeWorld", "FakeComm", "run_fake_cluster"]


class FakeWorld:
    def __init__(self, world_size: int, timeout: float = 60.0):
        self.n = world_size
        self.timeout = timeout
        self.lock = threading.Lock()
        self.barrier = threading.Barrier(world_size, timeout=timeout)
        self.slots: Dict[str, List] = {}
        self.results: Dict[str, object] = {}

    def collective(self, key: str, rank: int, payload,
                   combine: Callable[[List], object]):
        with self.lock:
            self.slots.setdefault(key, [None] * self.n)[rank] = payload
        self.barrier.wait()
        if rank == 0:
            self.results[key] = combine(self.slots[key])
        self.barrier.wait()
        out = self.results[key]
        self.barrier.wait()
        if rank == 0:
            with self.lock:
                self.slots.pop(key, None)
                self.results.pop(key, None)
        return out


class FakeComm(Comm):
    """Comm backed by a FakeWorld instead of torch.distributed."""

    def __init__(self, world: FakeWorld, rank: int,
                 device: Optional[torch.device] = None,
                 fail_at: Optional[int] = None):
        super().__init__(rank=rank, world_size=world.n, backend="fake",
                         device=device or torch.device("cpu"))
        self.world = world
        self._seq = 0
        self._fail_at = fail_at  # fault injection: die before op #fail_at

    def _next(self, op: str) -> str:
        self._seq += 1
        if self._fail_at is not None and self._seq >= self._fail_at:
            raise ConnectionError(
                f"[fake] rank {self.rank} injected failure at op {self._seq}")
        return f"{op}:{self._seq}"

    @property
    def is_distributed(self) -> bool:  # type: ignore[override]
        return True

    def all_reduce(self, t, async_op: bool = False):
        key = self._next("ar")
        out = self.world.collective(
            key, self.rank, t.detach().clone(),
            lambda xs: torch.stack(xs).sum(0))
        t.copy_(out)
        return _Done()

    def reduce_scatter(self, out, inp, async_op: bool = False):
        key = self._next("rs")
        total = self.world.collective(
            key, self.rank, inp.detach().clone(),
            lambda xs: torch.stack(xs).sum(0))
        k = out.numel()
        out.copy_(total[self.rank * k:(self.rank + 1) * k])
        return _Done()

    def all_gather(self, out, shard, async_op: bool = False):
        key = self._next("ag")
        full = self.world.collective(
            key, self.rank, shard.detach().clone(),
            lambda xs: torch.cat(xs))
        out.copy_(full)
        return _Done()

    def broadcast(self, t, src: int = 0, async_op: bool = False):
        key = self._next("bc")
        out = self.world.collective(key, self.rank, t.detach().clone(),
                                    lambda xs: xs[src])
        t.copy_(out)
        return _Done()

    def barrier(self) -> None:
        self.world.collective(self._next("b"), self.rank, None,
                              lambda xs: None)

    def supports_reduce_scatter(self) -> bool:
        return True


def run_fake_cluster(world_size: int, rank_fn: Callable[[FakeComm, int], object],
                     timeout: float = 120.0):
    """Run rank_fn(comm, rank) on N threads; returns list of results,
    re-raising the first rank exception."""
    world = FakeWorld(world_size, timeout=timeout)
    results: List = [None] * world_size
    errors: List = [None] * world_size

    def runner(r):
        try:
            results[r] = rank_fn(FakeComm(world, r), r)
        except BaseException as e:  # noqa: BLE001
            errors[r] = e
            world.barrier.abort()

    threads = [threading.Thread(target=runner, args=(r,), daemon=True)
               for r in range(world_size)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout)
    for e in errors:
        if e is not None and not isinstance(e, threading.BrokenBarrierError):
            raise e
    for e in errors:
        if e is not None:
            raise e
    return results
