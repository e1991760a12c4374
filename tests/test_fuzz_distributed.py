"""Property-based checks of the multi-GPU sharding helpers (no process
group needed: pure partition/range logic)."""
from hypothesis import given, settings, strategies as st

from lighthouse_amd.distributed import partition_sets, shard_range

SET = settings(max_examples=50, deadline=None)


@SET
@given(
    st.lists(st.integers(1, 512), min_size=0, max_size=200),
    st.integers(1, 8),
)
def test_partition_exact_cover(costs, world):
    parts = partition_sets(costs, world)
    assert len(parts) == world
    seen = sorted(i for p in parts for i in p)
    assert seen == list(range(len(costs)))  # every set exactly once
    for p in parts:
        assert p == sorted(p)


@SET
@given(
    st.lists(st.integers(1, 512), min_size=8, max_size=200),
    st.integers(2, 8),
)
def test_partition_balance(costs, world):
    # greedy bound: max load <= avg + max single cost
    parts = partition_sets(costs, world)
    loads = [sum(costs[i] for i in p) for p in parts]
    assert max(loads) <= sum(costs) / world + max(costs)


@given(st.integers(0, 5), st.integers(3, 20))
@SET
def test_shard_range_partition(log_world, log_n):
    world = 1 << log_world
    n = 1 << log_n
    if world > n:
        return
    covered = []
    for r in range(world):
        start, per = shard_range(n, r, world)
        assert per & (per - 1) == 0
        covered.append((start, start + per))
    covered.sort()
    assert covered[0][0] == 0 and covered[-1][1] == n
    for (a, b), (c, d) in zip(covered, covered[1:]):
        assert b == c  # contiguous, disjoint
