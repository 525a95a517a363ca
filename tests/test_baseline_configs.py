"""BASELINE.json configuration scenarios as tests (fast variants of
benchmarks/sweep.py: placements must be correct under each named config)."""
from __future__ import annotations

import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent / "benchmarks"))

import sweep  # noqa: E402


def test_config1_single_gpu_node():
    r = sweep.config1.__wrapped__() if hasattr(sweep.config1, "__wrapped__") \
        else sweep.config1()
    assert r["pods_per_sec"] > 0
    assert r["p50_ms"] < 100


def test_config2_memory_sharing_one_card():
    assert sweep.config2()["ok"]


def test_config3_whole_card_spread():
    assert sweep.config3()["ok"]


def test_config5_topology_adjacent():
    assert sweep.config5()["ok"]


def test_cpx_partitioned_node_model():
    """CPX partition mode: each OAM exposes 8 partitions of ~36 GB, so an
    8-OAM node enumerates 64 'cards'. Partitions of one OAM are closer
    (hop 1) than cross-OAM (hop 2): multi-card pods must stay intra-OAM and
    fractional accounting must respect the smaller per-partition memory."""
    import json

    from elastic_gpu_scheduler_amd.k8s.client import FakeKubeClient
    from elastic_gpu_scheduler_amd.k8s import objects as obj
    from elastic_gpu_scheduler_amd.scheduler.service import SchedulerRegistry
    from tests.conftest import make_pod

    GiB = 1024**3
    cards = [{"index": i, "core": 100, "memory_bytes": 36 * GiB,
              "partition": "CPX"} for i in range(64)]
    hops = [[0 if i == j else (1 if i // 8 == j // 8 else 2)
             for j in range(64)] for i in range(64)]
    client = FakeKubeClient()
    client.add_node({"metadata": {"name": "cpx", "annotations": {
        "elasticgpu.io/gpu-inventory": json.dumps({"cards": cards}),
        "elasticgpu.io/xgmi-topology": json.dumps({"hops": hops})}},
        "status": {}})
    sch = SchedulerRegistry(client).default

    # memory beyond one partition (37 GiB > 36) must be infeasible
    big = client.create_pod(make_pod("big", memory=37 * GiB))
    ok, failed = sch.assume(["cpx"], big)
    assert ok == [] and "cpx" in failed

    # a 4-partition pod lands inside ONE OAM (hop-1 set)
    quad = client.create_pod(make_pod("quad", per_container=[{"pgpu": 4}]))
    ok, _ = sch.assume(["cpx"], quad)
    assert ok == ["cpx"]
    sch.bind("cpx", quad)
    alloc = obj.parse_allocation(client.get_pod("default", "quad"))
    oams = {i // 8 for i in alloc[0]}
    assert len(alloc[0]) == 4 and len(oams) == 1, alloc

    # fractional pods fit within the 36 GiB partitions
    frac = client.create_pod(make_pod("frac", core=50, memory=18 * GiB))
    ok, _ = sch.assume(["cpx"], frac)
    assert ok == ["cpx"]
    sch.bind("cpx", frac)
    devs = sch.state.node_devices("cpx")
    assert len(devs) == 64
    assert any(d.mem_avail == 18 * GiB for d in devs)
