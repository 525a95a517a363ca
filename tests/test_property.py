"""Property-based tests (hypothesis): the placement search against the
brute-force oracle on random configurations, quantity parsing, codec
roundtrips, and allocator accounting invariants."""
from __future__ import annotations

import itertools

from hypothesis import given, settings, strategies as st

from elastic_gpu_scheduler_amd._native import core
from elastic_gpu_scheduler_amd.k8s import objects as obj
from elastic_gpu_scheduler_amd.utils.quantity import parse_memory_bytes
from tests.conftest import make_pod
from tests.test_search import brute_force_feasible, to_native

GiB = 1024**3

# random small cluster states: up to 5 cards with partial availability
card = st.tuples(st.integers(0, 100), st.integers(0, 288) .map(lambda g: g * GiB))
avail_strategy = st.lists(card, min_size=1, max_size=5)
unit = st.one_of(
    st.tuples(st.just("whole"), st.integers(1, 3), st.just(0)),
    st.tuples(st.just("frac"), st.integers(1, 100), st.integers(0, 300).map(lambda g: g * GiB)),
    st.tuples(st.just("none"), st.just(0), st.just(0)),
)
req_strategy = st.lists(unit, min_size=1, max_size=3)


@settings(max_examples=200, deadline=None)
@given(avail=avail_strategy, req=req_strategy)
def test_search_feasibility_matches_oracle(avail, req):
    # cards with partial core: the oracle treats 'whole' as needing a card
    # whose avail == total; mirror that by making totals == avail for frac
    # and keeping core_total=100 (so partially-used cards are not whole-free)
    devs = []
    for c, m in avail:
        devs.append(core.Device(100, c, 288 * GiB, m))
    units = to_native(avail, req)[1]
    feasible, opt, _ = core.search_placement(devs, units, "binpack", 0, [])
    # oracle's whole-free check: avail == (100, mem) and mem == 288GiB? The
    # native whole_free() is core_avail==core_total AND mem_avail==mem_total.
    def oracle():
        n = len(avail)

        def rec(state, i):
            if i == len(req):
                return True
            kind, a, b = req[i]
            if kind == "none":
                return rec(state, i + 1)
            if kind == "whole":
                free = [j for j in range(n)
                        if state[j][0] == 100 and state[j][1] == 288 * GiB]
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

    assert feasible == oracle(), (avail, req)


@settings(max_examples=200, deadline=None)
@given(n=st.integers(0, 10**15),
       suffix=st.sampled_from(["", "Ki", "Mi", "Gi", "Ti", "k", "M", "G"]))
def test_quantity_parse_consistent(n, suffix):
    mult = {"": None, "Ki": 1024, "Mi": 1024**2, "Gi": 1024**3,
            "Ti": 1024**4, "k": 10**3, "M": 10**6, "G": 10**9}[suffix]
    out = parse_memory_bytes(f"{n}{suffix}")
    if mult is not None:
        assert out == n * mult
    else:
        # bare: auto heuristic
        assert out == (n * GiB if 0 < n < 8192 else n)


@settings(max_examples=100, deadline=None)
@given(alloc=st.lists(st.lists(st.integers(0, 15), max_size=8), min_size=1,
                      max_size=4))
def test_annotation_codec_roundtrip(alloc):
    pod = make_pod("p", containers=len(alloc), core=10)
    annotated = obj.apply_allocation(pod, alloc)
    assert obj.parse_allocation(annotated) == alloc


@settings(max_examples=50, deadline=None)
@given(ops=st.lists(
    st.tuples(st.sampled_from(["alloc", "forget"]), st.integers(0, 9),
              st.integers(1, 60), st.integers(0, 100)),
    max_size=40))
def test_allocator_accounting_balances(ops):
    """Random alloc/forget interleavings: availability never goes negative
    and forgetting everything restores the pristine state."""
    c = core.ClusterState("binpack", 0, 0)
    c.add_node("n", [core.Device(100, 100, 288 * GiB, 288 * GiB)
                     for _ in range(4)], [])
    live = set()
    for kind, pid, core_pct, mem_g in ops:
        uid = f"p{pid}"
        if kind == "alloc" and uid not in live:
            try:
                c.allocate("n", uid, [core.GPUUnit(0, core_pct, mem_g * GiB)])
                live.add(uid)
            except RuntimeError:
                pass
        elif kind == "forget" and uid in live:
            c.forget_pod(uid)
            live.discard(uid)
        for d in c.node_devices("n"):
            assert d.core_avail >= 0 and d.mem_avail >= 0
    for uid in list(live):
        c.forget_pod(uid)
    for d in c.node_devices("n"):
        assert d.core_avail == 100 and d.mem_avail == 288 * GiB


@settings(max_examples=100, deadline=None)
@given(payload=st.recursive(
    st.one_of(st.none(), st.booleans(), st.integers(-2**40, 2**40),
              st.text(max_size=20)),
    lambda children: st.one_of(
        st.lists(children, max_size=4),
        st.dictionaries(st.text(max_size=8), children, max_size=4)),
    max_leaves=20))
def test_native_json_codec_matches_python(payload):
    import json

    encoded = json.dumps(payload)
    assert json.loads(core.json_roundtrip(encoded)) == payload


# ---------------------------------------------------------------------------
# Stateful: random op sequences against a pure-Python accounting model.
# A handful of repeating shapes FORCES shape-cache sharing across pods
# (the r2 large-cluster memo); the model proves the cache can never leak
# stale feasibility into a commit (the reference's shape-cache sharing
# double-books, allocate.go:30-33).

_SHAPES = [
    [core.GPUUnit(0, 30, 16 * GiB)],
    [core.GPUUnit(0, 60, 64 * GiB)],
    [core.GPUUnit(gpu_count=1)],
    [core.GPUUnit(gpu_count=2)],
]


def _model_feasible(model_devs, shape):
    u = shape[0]
    if u.gpu_count > 0:
        free = sum(1 for c, m in model_devs
                   if c == 100 and m == 288 * GiB)
        return free >= u.gpu_count
    return any(c >= u.core and m >= u.memory for c, m in model_devs)


_op = st.tuples(st.sampled_from(["assume", "allocate", "forget"]),
                st.integers(0, 15),   # pod id
                st.integers(0, 1),    # node
                st.integers(0, len(_SHAPES) - 1))


@settings(max_examples=120, deadline=None)
@given(ops=st.lists(_op, min_size=1, max_size=60))
def test_cluster_state_matches_python_model(ops):
    state = core.ClusterState("binpack", 0, 2)
    nodes = ["m0", "m1"]
    for n in nodes:
        state.add_node(n, [core.Device() for _ in range(3)], [])
    # model: node -> [(core_avail, mem_avail)]; pod -> (node, shape, cards)
    model = {n: [(100, 288 * GiB)] * 3 for n in nodes}
    committed = {}
    assumed_ok = set()  # uids holding a cached feasible placement

    def apply(node, cards, shape, sign):
        u = shape[0]
        for c in cards:
            ca, ma = model[node][c]
            if u.gpu_count > 0:
                model[node][c] = ((100, 288 * GiB) if sign < 0 else (0, 0))
            else:
                model[node][c] = (ca - sign * u.core,
                                  ma - sign * u.memory)

    for op, pid, nid, sid in ops:
        uid = f"pod-{pid}"
        node = nodes[nid]
        shape = _SHAPES[sid]
        if op == "assume":
            verdicts = state.assume([node], uid, shape)
            if uid not in committed and uid not in assumed_ok:
                # no cached per-uid placement exists and assume commits
                # nothing, so the native verdict must MATCH the model
                expected = _model_feasible(model[node], shape)
                assert (verdicts[0] == 0) == expected, (
                    verdicts, shape, model[node])
            if verdicts[0] == 0:
                assumed_ok.add(uid)
        elif op == "allocate":
            if uid in committed:
                continue  # idempotent re-allocate returns old placement
            try:
                opt = state.allocate(node, uid, shape)
            except RuntimeError:
                # must ONLY fail when the model agrees it cannot fit
                assert not _model_feasible(model[node], shape), (
                    model[node], shape)
                continue
            cards = [c for a in opt.allocated for c in a]
            u = shape[0]
            # returned placement must fit the MODEL's pre-state
            if u.gpu_count > 0:
                assert len(set(cards)) == u.gpu_count
                for c in cards:
                    assert model[node][c] == (100, 288 * GiB), \
                        "double-booked whole card"
            else:
                (ca, ma) = model[node][cards[0]]
                assert ca >= u.core and ma >= u.memory, "over-committed"
            apply(node, cards, shape, +1)
            committed[uid] = (node, shape, cards)
        else:  # forget
            state.forget_pod(uid)
            # forget routes via the pod->node map, which only exists after
            # allocate: an assumed-but-never-bound pod keeps its per-node
            # cached placement until the TTL sweep (harmless in production:
            # pod UIDs are never reused with a different spec) — so the
            # model's "uid holds a cached placement" flag stays set.
            if uid in committed:
                assumed_ok.discard(uid)
                node_c, shape_c, cards = committed.pop(uid)
                apply(node_c, cards, shape_c, -1)

    # final accounting must match the model exactly
    for n in nodes:
        devs = state.node_devices(n)
        for i, d in enumerate(devs):
            assert (d.core_avail, d.mem_avail) == model[n][i], (
                n, i, (d.core_avail, d.mem_avail), model[n][i])
