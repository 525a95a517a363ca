"""Native core allocator: feasibility, sharing, whole-card, idempotency,
and the concurrency-safety properties the reference lacks."""
from __future__ import annotations

import pytest

from elastic_gpu_scheduler_amd._native import core

GiB = 1024**3


def devices(n=8, mem=288 * GiB):
    return [core.Device(100, 100, mem, mem) for _ in range(n)]


def cs(policy="binpack", seed=0, threads=0):
    return core.ClusterState(policy, seed, threads)


def frac(core_pct=0, mem=0):
    return [core.GPUUnit(0, core_pct, mem)]


def whole(n):
    return [core.GPUUnit(n, 0, 0)]


class TestFractional:
    def test_share_one_card(self):
        c = cs()
        c.add_node("n", devices(1), [])
        for i in range(4):
            assert c.assume(["n"], f"p{i}", frac(25, 64 * GiB)) == [0]
            c.allocate("n", f"p{i}", frac(25, 64 * GiB))
        d = c.node_devices("n")[0]
        assert d.core_avail == 0
        assert d.mem_avail == (288 - 256) * GiB
        # card is full now
        assert c.assume(["n"], "p5", frac(25, 64 * GiB)) == [1]

    def test_memory_only_request(self):
        c = cs()
        c.add_node("n", devices(1), [])
        assert c.assume(["n"], "p", frac(0, 200 * GiB)) == [0]
        opt = c.allocate("n", "p", frac(0, 200 * GiB))
        assert opt.allocated == [[0]]
        assert c.assume(["n"], "q", frac(0, 100 * GiB)) == [1]  # doesn't fit

    def test_no_gpu_container_gets_empty(self):
        c = cs()
        c.add_node("n", devices(2), [])
        req = [core.GPUUnit(0, 0, 0), core.GPUUnit(0, 30, GiB)]
        c.assume(["n"], "p", req)
        opt = c.allocate("n", "p", req)
        assert opt.allocated[0] == []
        assert len(opt.allocated[1]) == 1

    def test_infeasible_when_fragmented(self):
        # 2 cards at 60% used each: a 50% request fits nowhere.
        c = cs()
        c.add_node("n", devices(2), [])
        for i in range(2):
            c.allocate("n", f"pre{i}", frac(60, GiB))
        assert c.assume(["n"], "p", frac(50, GiB)) == [1]


class TestWholeCard:
    def test_exclusive(self):
        c = cs()
        c.add_node("n", devices(2), [])
        c.allocate("n", "p", whole(1))
        # the taken card is fully zeroed; a fractional pod fits on the other
        avail = sorted((d.core_avail, d.mem_avail) for d in c.node_devices("n"))
        assert avail[0] == (0, 0)
        assert c.assume(["n"], "q", whole(2)) == [1]
        assert c.assume(["n"], "r", whole(1)) == [0]

    def test_whole_requires_fully_free(self):
        c = cs()
        c.add_node("n", devices(1), [])
        c.allocate("n", "p", frac(1, GiB))
        assert c.assume(["n"], "q", whole(1)) == [1]

    def test_multi_card(self):
        c = cs()
        c.add_node("n", devices(8), [])
        opt = c.allocate("n", "p", whole(4))
        assert len(opt.allocated[0]) == 4
        assert len(set(opt.allocated[0])) == 4

    def test_mixed_containers(self):
        c = cs()
        c.add_node("n", devices(4), [])
        req = [core.GPUUnit(2, 0, 0), core.GPUUnit(0, 50, 10 * GiB)]
        opt = c.allocate("n", "p", req)
        assert len(opt.allocated[0]) == 2
        assert len(opt.allocated[1]) == 1
        assert not set(opt.allocated[0]) & set(opt.allocated[1])


class TestLifecycle:
    def test_forget_restores(self):
        c = cs()
        c.add_node("n", devices(1), [])
        c.allocate("n", "p", frac(40, 10 * GiB))
        c.forget_pod("p")
        d = c.node_devices("n")[0]
        assert d.core_avail == 100 and d.mem_avail == 288 * GiB

    def test_allocate_idempotent(self):
        c = cs()
        c.add_node("n", devices(1), [])
        o1 = c.allocate("n", "p", frac(40, 10 * GiB))
        o2 = c.allocate("n", "p", frac(40, 10 * GiB))
        assert o1.allocated == o2.allocated
        assert c.node_devices("n")[0].core_avail == 60  # charged once

    def test_forget_unknown_is_noop(self):
        c = cs()
        c.add_node("n", devices(1), [])
        c.forget_pod("nope")

    def test_add_pod_replay(self):
        c = cs()
        c.add_node("n", devices(2), [])
        opt = core.GPUOption()
        opt.allocated = [[1]]
        c.add_pod("n", "p", frac(30, 5 * GiB), opt)
        assert c.node_devices("n")[1].core_avail == 70
        # replay is idempotent
        c.add_pod("n", "p", frac(30, 5 * GiB), opt)
        assert c.node_devices("n")[1].core_avail == 70
        # forget via cluster-level uid map
        c.forget_pod("p")
        assert c.node_devices("n")[1].core_avail == 100

    def test_add_pod_rejects_overbooked_replay(self):
        c = cs()
        c.add_node("n", devices(1), [])
        c.allocate("n", "a", frac(80, GiB))
        opt = core.GPUOption()
        opt.allocated = [[0]]
        with pytest.raises(RuntimeError):
            c.add_pod("n", "b", frac(80, GiB), opt)

    def test_allocate_infeasible_raises(self):
        c = cs()
        c.add_node("n", devices(1), [])
        c.allocate("n", "p", whole(1))
        with pytest.raises(RuntimeError):
            c.allocate("n", "q", whole(1))


class TestAssumeCache:
    def test_same_shape_pods_get_independent_options(self):
        """The reference keys its assume cache by request-shape hash
        (allocate.go:30-33), so two identical pods share one cached option;
        ours is per-UID: both must be placeable."""
        c = cs()
        c.add_node("n", devices(1), [])
        r = frac(60, 100 * GiB)
        assert c.assume(["n"], "p1", r) == [0]
        assert c.assume(["n"], "p2", r) == [0]
        c.allocate("n", "p1", r)
        # p2's assumed option must re-validate: only 40 core left -> infeasible
        with pytest.raises(RuntimeError):
            c.allocate("n", "p2", r)

    def test_score_without_assume(self):
        """Reference Score nil-derefs on a cache miss (node.go:78-84)."""
        c = cs()
        c.add_node("n", devices(1), [])
        scores = c.score(["n"], "fresh-pod", frac(30, GiB))
        assert 0.0 <= scores[0] <= 10.0

    def test_unknown_node_verdict(self):
        c = cs()
        assert c.assume(["ghost"], "p", frac(10, 0)) == [2]


class TestDeterminism:
    def test_same_inputs_same_placement(self):
        for policy in ("binpack", "spread", "random"):
            opts = []
            for _ in range(3):
                c = cs(policy, seed=7)
                c.add_node("n", devices(8), [])
                c.allocate("n", "warm", frac(50, 10 * GiB))
                opts.append(c.allocate("n", "p", frac(30, 5 * GiB)).allocated)
            assert opts[0] == opts[1] == opts[2], policy


def test_shape_cache_never_double_books():
    """The per-shape search memo (large-cluster fast path) must stay a pure
    MEMO: two same-shaped pods may receive the same cached proposal, but
    commit re-validates — the second bind must fail or land elsewhere, and
    any state change invalidates the cached verdict (generation bump)."""
    from elastic_gpu_scheduler_amd._native import core

    state = core.ClusterState("binpack", 0, 2)
    state.add_node("n", [core.Device(100, 100, GiB, GiB)], [])
    unit = [core.GPUUnit(0, 60, 0)]  # 60% core: only one fits per card
    assert state.assume(["n"], "pod-a", unit) == [0]
    assert state.assume(["n"], "pod-b", unit) == [0]  # same shape, cache hit
    opt = state.allocate("n", "pod-a", unit)
    assert opt.allocated[0] == [0]
    # pod-b's cached proposal no longer fits; allocate must refuse
    import pytest as _pytest
    with _pytest.raises(Exception):
        state.allocate("n", "pod-b", unit)
    # generation bumped: a fresh same-shaped assume now reports infeasible
    assert state.assume(["n"], "pod-c", unit) == [1]
    # release invalidates again: feasible once more
    state.forget_pod("pod-a")
    assert state.assume(["n"], "pod-d", unit) == [0]
    assert state.allocate("n", "pod-d", unit).allocated[0] == [0]
