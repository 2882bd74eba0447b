"""Shard-planner tests (capability C2): the reference's two policies."""

import torch

from rocps.placement import (greedy_load_balancing_strategy, plan_shards,
                             replica_device_setter, round_robin_strategy)


def _params(sizes):
    return [(f"p{i}", torch.zeros(s)) for i, s in enumerate(sizes)]


def test_round_robin_golden():
    plan = plan_shards(_params([10, 20, 30, 40, 50]), 3, "round_robin")
    assert [plan.shard_of(f"p{i}") for i in range(5)] == [0, 1, 2, 0, 1]


def test_greedy_by_bytes_golden():
    # sizes (elements→bytes×4): 100, 10, 10, 90, 10
    plan = plan_shards(_params([100, 10, 10, 90, 10]), 2, "greedy")
    # p0→shard0(100); p1→shard1(10); p2→shard1(20); p3→shard1(110); p4→shard0(110)
    assert plan.shard_of("p0") == 0
    assert plan.shard_of("p1") == 1
    assert plan.shard_of("p2") == 1
    assert plan.shard_of("p3") == 1
    assert plan.shard_of("p4") == 0
    assert plan.shard_bytes[0] == plan.shard_bytes[1] == 110 * 4


def test_greedy_balance_quality():
    sizes = [1000, 500, 500, 250, 250, 250, 250]
    plan = plan_shards(_params(sizes), 2, "greedy")
    a, b = plan.shard_bytes
    assert abs(a - b) <= 4 * max(sizes) // 2


def test_strategies_direct():
    rr = round_robin_strategy(4)
    assert [rr("x", 1) for _ in range(6)] == [0, 1, 2, 3, 0, 1]
    g = greedy_load_balancing_strategy(2)
    assert g("a", 100) in (0, 1)


def test_replica_device_setter_shim():
    cs_like = None
    setter = replica_device_setter(ps_tasks=3)
    t = torch.zeros(10)
    assert [setter(f"v{i}", t) for i in range(4)] == [0, 1, 2, 0]
    # no ps → everything on shard 0
    setter0 = replica_device_setter(ps_tasks=0)
    assert setter0("v", t) == 0


def test_placement_fuzz_invariants():
    """Property fuzz: both strategies assign every tensor exactly once;
    greedy's max shard load never exceeds min load + the largest
    tensor (the classic LPT-greedy bound for online assignment)."""
    from hypothesis import given, settings, strategies as st

    @settings(deadline=None, max_examples=40)
    @given(
        sizes=st.lists(st.integers(1, 10_000), min_size=1, max_size=40),
        shards=st.integers(1, 8),
        strategy=st.sampled_from(["round_robin", "greedy"]),
    )
    def check(sizes, shards, strategy):
        named = [(f"v{i}", n) for i, n in enumerate(sizes)]
        plan = plan_shards(named, shards, strategy=strategy)
        assigned = [plan.shard_of(f"v{i}") for i in range(len(sizes))]
        assert all(0 <= s < shards for s in assigned)
        per_names = [plan.names_on(s) for s in range(shards)]
        flat = [n for ns in per_names for n in ns]
        assert sorted(flat) == sorted(n for n, _ in named)  # exactly once
        if strategy == "greedy":
            loads = [0] * shards
            for (n, b), s in zip(named, assigned):
                loads[s] += b
            assert max(loads) <= min(loads) + max(sizes)

    check()
