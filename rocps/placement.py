"""Parameter→PS-shard placement — capability C2 of SURVEY.md §2b.

The reference uses ``tf.train.replica_device_setter`` to place each
variable on ``/job:ps/task:i`` round-robin, or balanced by bytes via
``tf.train.GreedyLoadBalancingStrategy(num_ps, byte_size_load_fn)``.

Here a *shard plan* maps each named parameter tensor of a model to a PS
shard index. Sync mode uses it to slice the flat bucket space so each
rank's reduce-scattered segment is a balanced co-located shard; async
mode uses it to pick which host-DRAM PS shard a tensor lives on.
"""

from __future__ import annotations

import heapq
from typing import Callable, Dict, Iterable, List, Tuple

__all__ = [
    "round_robin_strategy", "greedy_load_balancing_strategy",
    "plan_shards", "replica_device_setter", "ShardPlan",
]


class ShardPlan:
    """name -> shard index, plus per-shard tensors/bytes accounting."""

    def __init__(self, num_shards: int):
        self.num_shards = num_shards
        self.assignment: Dict[str, int] = {}
        self.shard_bytes: List[int] = [0] * num_shards

    def assign(self, name: str, shard: int, nbytes: int) -> None:
        self.assignment[name] = shard
        self.shard_bytes[shard] += nbytes

    def shard_of(self, name: str) -> int:
        return self.assignment[name]

    def names_on(self, shard: int) -> List[str]:
        return [n for n, s in self.assignment.items() if s == shard]

    def __repr__(self) -> str:
        return (f"ShardPlan(num_shards={self.num_shards}, "
                f"tensors={len(self.assignment)}, bytes={self.shard_bytes})")


def round_robin_strategy(num_shards: int) -> Callable[[str, int], int]:
    """The reference's default: variables assigned to ps tasks in turn."""
    state = {"next": 0}

    def choose(name: str, nbytes: int) -> int:
        s = state["next"]
        state["next"] = (s + 1) % num_shards
        return s

    return choose


def greedy_load_balancing_strategy(num_shards: int) -> Callable[[str, int], int]:
    """Greedy by bytes: each tensor goes to the currently-lightest shard.

    Mirrors ``tf.train.GreedyLoadBalancingStrategy`` with
    ``byte_size_load_fn``.
    """
    heap: List[Tuple[int, int]] = [(0, i) for i in range(num_shards)]
    heapq.heapify(heap)

    def choose(name: str, nbytes: int) -> int:
        load, shard = heapq.heappop(heap)
        heapq.heappush(heap, (load + nbytes, shard))
        return shard

    return choose


def plan_shards(named_params: Iterable[Tuple[str, "object"]],
                num_shards: int,
                strategy: str = "round_robin") -> ShardPlan:
    """Build a ShardPlan for (name, tensor) pairs.

    ``strategy``: "round_robin" (reference default) or "greedy" (bytes).
    """
    if num_shards <= 0:
        raise ValueError("num_shards must be positive")
    if strategy == "round_robin":
        choose = round_robin_strategy(num_shards)
    elif strategy in ("greedy", "greedy_load_balancing"):
        choose = greedy_load_balancing_strategy(num_shards)
    else:
        raise ValueError(f"unknown placement strategy {strategy!r}")
    plan = ShardPlan(num_shards)
    for name, p in named_params:
        nbytes = p.numel() * p.element_size() if hasattr(p, "numel") else int(p)
        plan.assign(name, choose(name, nbytes), nbytes)
    return plan


def replica_device_setter(ps_tasks: int = 0, cluster=None,
                          strategy: str = "round_robin"):
    """API-compat shim for the reference's ``tf.train.replica_device_setter``.

    Returns a callable mapping (name, tensor) -> shard index using the
    same two policies the reference exposes. With a ClusterSpec, the
    shard count is the number of ps tasks.
    """
    if cluster is not None:
        ps_tasks = cluster.num_tasks("ps") if "ps" in cluster.jobs else 0
    if ps_tasks <= 0:
        return lambda name, p: 0
    choose = (round_robin_strategy(ps_tasks) if strategy == "round_robin"
              else greedy_load_balancing_strategy(ps_tasks))

    def setter(name: str, p) -> int:
        nbytes = p.numel() * p.element_size() if hasattr(p, "numel") else int(p)
        return choose(name, nbytes)

    return setter
