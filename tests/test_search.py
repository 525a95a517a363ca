"""Placement search: correctness vs a Python brute-force oracle on small
configurations, plus budget/pruning behavior."""
from __future__ import annotations

import itertools

from elastic_gpu_scheduler_amd._native import core

GiB = 1024**3


def brute_force_feasible(avail, req):
    """Oracle: does ANY assignment exist? avail = [(core, mem)], req =
    [(kind, a, b)] with kind 'whole' (a=count) or 'frac' (a=core, b=mem)."""
    n = len(avail)

    def rec(state, i):
        if i == len(req):
            return True
        kind, a, b = req[i]
        if kind == "none":
            return rec(state, i + 1)
        if kind == "whole":
            free = [j for j in range(n)
                    if state[j] == avail[j] and state[j][0] == 100]
            for combo in itertools.combinations(free, a):
                s2 = list(state)
                for j in combo:
                    s2[j] = (0, 0)
                if rec(s2, i + 1):
                    return True
            return False
        for j in range(n):
            cj, mj = state[j]
            if cj >= a and mj >= b:
                s2 = list(state)
                s2[j] = (cj - a, mj - b)
                if rec(s2, i + 1):
                    return True
        return False

    return rec(list(avail), 0)


def to_native(avail, req):
    devs = []
    for c, m in avail:
        devs.append(core.Device(100, c, 288 * GiB, m))
    units = []
    for kind, a, b in req:
        if kind == "whole":
            units.append(core.GPUUnit(a, 0, 0))
        elif kind == "frac":
            units.append(core.GPUUnit(0, a, b))
        else:
            units.append(core.GPUUnit(0, 0, 0))
    return devs, units


CASES = [
    # (avail per card, request per container)
    ([(100, 288 * GiB)] * 4, [("frac", 30, 10 * GiB)]),
    ([(100, 288 * GiB)] * 4, [("whole", 2, 0), ("frac", 50, GiB)]),
    ([(50, 100 * GiB), (100, 288 * GiB)], [("whole", 1, 0)]),
    ([(50, 100 * GiB), (40, 20 * GiB)], [("frac", 45, 50 * GiB)]),
    ([(10, GiB)] * 3, [("frac", 20, 0)]),                      # infeasible
    ([(100, 288 * GiB)] * 2, [("whole", 3, 0)]),               # infeasible
    ([(100, 288 * GiB)] * 3,
     [("frac", 60, 0), ("frac", 60, 0), ("frac", 60, 0), ("frac", 60, 0)]),
    ([(100, 288 * GiB)] * 2, [("none", 0, 0), ("frac", 10, GiB)]),
    ([(30, 10 * GiB), (30, 10 * GiB), (100, 288 * GiB)],
     [("frac", 25, 5 * GiB), ("whole", 1, 0)]),
    ([(100, 64 * GiB)] * 4, [("frac", 0, 65 * GiB)]),          # mem infeasible
]


def test_search_matches_oracle_feasibility():
    for avail, req in CASES:
        devs, units = to_native(avail, req)
        for policy in ("binpack", "spread"):
            feasible, opt, _ = core.search_placement(devs, units, policy, 0, [])
            assert feasible == brute_force_feasible(avail, req), (avail, req, policy)


def test_search_respects_capacity():
    """Applying the returned option must never overdraw any card."""
    for avail, req in CASES:
        devs, units = to_native(avail, req)
        feasible, opt, _ = core.search_placement(devs, units, "binpack", 0, [])
        if not feasible:
            continue
        use = [[0, 0] for _ in avail]
        for c, cards in enumerate(opt.allocated):
            kind, a, b = req[c]
            for j in cards:
                if kind == "whole":
                    use[j][0] += 100
                    use[j][1] += avail[j][1]
                else:
                    use[j][0] += a
                    use[j][1] += b
        for j, (c_used, m_used) in enumerate(use):
            assert c_used <= avail[j][0]
            assert m_used <= avail[j][1]


def test_leaf_budget_bounds_work():
    """A worst-case shape (many containers x many cards) stays within the
    deterministic leaf budget instead of exploding combinatorially."""
    devs = [core.Device(100, 100, 288 * GiB, 288 * GiB) for _ in range(16)]
    units = [core.GPUUnit(0, 5, GiB) for _ in range(10)]
    feasible, opt, leaves = core.search_placement(devs, units, "binpack", 0, [])
    assert feasible
    assert leaves <= 4096


def test_whole_card_picks_topology_best_not_first_free():
    """With card 0 free but isolated, and cards 2,3 linked, a 2-card pod must
    take the linked pair (the reference would take the first 2 free:
    gpu.go:96-108)."""
    hops = [
        [0, 3, 3, 3],
        [3, 0, 3, 3],
        [3, 3, 0, 1],
        [3, 3, 1, 0],
    ]
    devs = [core.Device(100, 100, 288 * GiB, 288 * GiB) for _ in range(4)]
    units = [core.GPUUnit(2, 0, 0)]
    feasible, opt, _ = core.search_placement(devs, units, "binpack", 0, hops)
    assert feasible
    assert sorted(opt.allocated[0]) == [2, 3]


def test_combined_constraints_hive_plus_spread():
    """Whole-card hive locality + spread-containers + fractional mix, all at
    once, under the leaf budget (the combination the demo's hardest pod
    exercises)."""
    hops = [[0 if i == j else (1 if (i < 8) == (j < 8) else 3)
             for j in range(16)] for i in range(16)]
    devs = [core.Device(100, 100, 288 * GiB, 288 * GiB) for _ in range(16)]
    units = [core.GPUUnit(6, 0, 0), core.GPUUnit(0, 40, 8 * GiB),
             core.GPUUnit(0, 40, 8 * GiB)]
    feasible, opt, leaves = core.search_placement(devs, units, "binpack", 0,
                                                  hops, True)
    assert feasible and leaves <= 4096
    assert len({i < 8 for i in opt.allocated[0]}) == 1  # in one hive
    all_cards = [c for a in opt.allocated for c in a]
    assert len(all_cards) == len(set(all_cards))  # spread honoured


def test_huge_subset_space_stays_bounded():
    devs = [core.Device(100, 100, 288 * GiB, 288 * GiB) for _ in range(20)]
    feasible, opt, leaves = core.search_placement(
        devs, [core.GPUUnit(10, 0, 0)], "binpack", 0, [])
    assert feasible and len(set(opt.allocated[0])) == 10
    assert leaves <= 4096  # C(20,10)=184,756 raw combinations


def test_doomed_last_container_fails_fast():
    """a multi-container pod whose LAST container can never fit
    (even on an empty node) must not explore every placement of the earlier
    containers while holding the node mutex — the pre-check rejects it with
    zero leaf evaluations."""
    import time

    devs = [core.Device(100, 100, 288 * GiB, 288 * GiB) for _ in range(64)]
    # Non-uniform topology disables the symmetric-branch dedupe.
    hops = [[0 if i == j else 1 + ((i + j) % 3) for j in range(64)]
            for i in range(64)]
    units = [core.GPUUnit(0, 10, GiB) for _ in range(6)]
    units.append(core.GPUUnit(0, 10, 10_000 * GiB))  # never fits any card
    t0 = time.monotonic()
    feasible, _, leaves = core.search_placement(devs, units, "binpack", 0, hops)
    dt = time.monotonic() - t0
    assert not feasible
    assert leaves == 0
    assert dt < 0.1  # fail-fast, not a cards^containers walk


def test_oversized_whole_card_ask_fails_fast():
    devs = [core.Device(100, 100, 288 * GiB, 288 * GiB) for _ in range(8)]
    feasible, _, leaves = core.search_placement(
        devs, [core.GPUUnit(gpu_count=9)], "binpack")
    assert not feasible and leaves == 0


def test_zero_capacity_placeholder_never_scheduled():
    """Zero-capacity devices (sick-card placeholders) are skipped by both
    whole-card and fractional placement."""
    devs = [core.Device(100, 100, 288 * GiB, 288 * GiB),
            core.Device(0, 0, 0, 0),
            core.Device(100, 100, 288 * GiB, 288 * GiB)]
    feasible, opt, _ = core.search_placement(
        devs, [core.GPUUnit(gpu_count=2)], "binpack")
    assert feasible and sorted(opt.allocated[0]) == [0, 2]
    feasible, opt, _ = core.search_placement(
        devs, [core.GPUUnit(0, 30, GiB), core.GPUUnit(0, 30, GiB)], "spread")
    assert feasible
    assert all(c != 1 for a in opt.allocated for c in a)
    # And a request needing more cards than the healthy count is infeasible.
    feasible, _, _ = core.search_placement(
        devs, [core.GPUUnit(gpu_count=3)], "binpack")
    assert not feasible


def test_bare_auto_threshold_single_source():
    """The C++ fast path's bare-number GiB/bytes threshold is pushed from
    utils/quantity.py at import — one definition, no per-path drift."""
    from elastic_gpu_scheduler_amd.utils import quantity

    assert core.get_bare_auto_gib_threshold() == quantity.BARE_AUTO_GIB_THRESHOLD


# ---------------------------------------------------------------------------
# Fragmentation-proof whole-card subset selection (the
# r1 lexicographic-prefix candidate cap could hide the minimum-hop set on a
# fragmented CPX-64 node). Three fragmentation patterns, oracle-checked.


def _cpx_hops(n=64, hive=8):
    """CPX 64-partition topology: 8 hives x 8; intra-hive 1 hop, cross 3."""
    return [[0 if i == j else (1 if i // hive == j // hive else 3)
             for j in range(n)] for i in range(n)]


def _cpx_devices(free_indexes, n=64):
    devs = []
    for i in range(n):
        if i in free_indexes:
            devs.append(core.Device(100, 100, 288 * GiB, 288 * GiB))
        else:
            devs.append(core.Device(100, 0, 288 * GiB, 0))  # occupied
    return devs


def test_fragmented_cpx_high_index_hive_is_found():
    """Pattern 1: the ONLY fully-free hive is the LAST one — exactly what
    the r1 lexicographic prefix cap could never reach. k=8 must land on it."""
    free = set()
    for h in range(7):          # hives 0..6: 5 free cards each (fragmented)
        free.update(range(h * 8, h * 8 + 5))
    free.update(range(56, 64))  # hive 7: fully free
    devs = _cpx_devices(free)
    feasible, opt, _ = core.search_placement(
        devs, [core.GPUUnit(gpu_count=8)], "binpack", 0, _cpx_hops())
    assert feasible
    assert sorted(opt.allocated[0]) == list(range(56, 64))


def test_fragmented_cpx_small_set_on_high_hive():
    """Pattern 2: k=4; the only hive with 4 free cards is hive 7 (indexes
    56-59). Every other hive offers 3 — a 4-set there costs 3+3*3=... more.
    C(25,4)=12,650 > exhaustive cap, so this exercises the greedy path."""
    free = set()
    for h in range(7):
        free.update(range(h * 8, h * 8 + 3))  # 3 free per low hive
    free.update(range(56, 60))                # hive 7: 4 free
    devs = _cpx_devices(free)
    feasible, opt, _ = core.search_placement(
        devs, [core.GPUUnit(gpu_count=4)], "binpack", 0, _cpx_hops())
    assert feasible
    assert sorted(opt.allocated[0]) == [56, 57, 58, 59]


def test_fragmented_cpx_mixed_set_matches_bruteforce_oracle():
    """Pattern 3: no hive can satisfy k=5 alone -> the optimum mixes hives.
    The chosen set's pairwise hop cost must equal the brute-force minimum
    over all C(20,5)=15,504 subsets."""
    import itertools

    free = sorted(
        list(range(16, 20)) +    # hive 2: 4 free
        list(range(40, 44)) +    # hive 5: 4 free
        [0, 1, 8, 9, 24, 25, 32, 33, 48, 49, 56, 57])  # 2 per other hive
    assert len(free) == 20
    hops = _cpx_hops()

    def cost(cards):
        return sum(hops[a][b] for a, b in itertools.combinations(cards, 2))

    oracle = min(cost(s) for s in itertools.combinations(free, 5))
    devs = _cpx_devices(set(free))
    feasible, opt, _ = core.search_placement(
        devs, [core.GPUUnit(gpu_count=5)], "binpack", 0, hops)
    assert feasible
    chosen = sorted(opt.allocated[0])
    assert cost(chosen) == oracle, (chosen, cost(chosen), oracle)


def test_subset_selection_is_deterministic():
    free = set(range(0, 64, 2))  # 32 free cards, alternating
    devs = _cpx_devices(free)
    results = set()
    for _ in range(3):
        _, opt, _ = core.search_placement(
            devs, [core.GPUUnit(gpu_count=6)], "binpack", 0, _cpx_hops())
        results.add(tuple(sorted(opt.allocated[0])))
    assert len(results) == 1
