"""Extender preemption verb: minimal victim sets that make the pod fit."""
from __future__ import annotations

from elastic_gpu_scheduler_amd.k8s import objects as obj
from elastic_gpu_scheduler_amd.scheduler.service import GPUUnitScheduler
from elastic_gpu_scheduler_amd.k8s.client import FakeKubeClient
from tests.conftest import make_node, make_pod

GiB = 1024**3


def setup_full_node(cards=2):
    client = FakeKubeClient()
    client.add_node(make_node("n1", cards=cards))
    sch = GPUUnitScheduler(client)
    victims = []
    for i in range(cards):
        p = client.create_pod(make_pod(f"v{i}", per_container=[{"pgpu": 1}]))
        sch.assume(["n1"], p)
        sch.bind("n1", p)
        victims.append(obj.pod_uid(client.get_pod("default", f"v{i}")))
    return client, sch, victims


def test_preemption_minimal_victims():
    client, sch, victims = setup_full_node(cards=2)
    pod = client.create_pod(make_pod("pre", per_container=[{"pgpu": 1}]))
    result = sch.process_preemption(pod, {"n1": victims})
    # one whole card wanted: evicting ONE victim suffices
    assert "n1" in result
    assert len(result["n1"]) == 1
    assert result["n1"][0] in victims


def test_preemption_needs_all_victims():
    client, sch, victims = setup_full_node(cards=2)
    pod = client.create_pod(make_pod("pre", per_container=[{"pgpu": 2}]))
    result = sch.process_preemption(pod, {"n1": victims})
    assert sorted(result["n1"]) == sorted(victims)


def test_preemption_infeasible_node_omitted():
    client, sch, victims = setup_full_node(cards=2)
    pod = client.create_pod(make_pod("pre", per_container=[{"pgpu": 4}]))
    result = sch.process_preemption(pod, {"n1": victims})
    assert result == {}


def test_preemption_zero_victims_when_already_fits():
    client = FakeKubeClient()
    client.add_node(make_node("n1"))
    sch = GPUUnitScheduler(client)
    v = client.create_pod(make_pod("v0", core=10, memory=GiB))
    sch.assume(["n1"], v)
    sch.bind("n1", v)
    pod = client.create_pod(make_pod("pre", core=10, memory=GiB))
    result = sch.process_preemption(
        pod, {"n1": [obj.pod_uid(client.get_pod("default", "v0"))]})
    assert result == {"n1": []}  # fits without evicting anyone


def test_preemption_http_handler(cluster, extender):
    client, registry, app = cluster
    sch = registry.default
    victims = []
    for i in range(8):
        p = client.create_pod(make_pod(f"v{i}", per_container=[{"pgpu": 1}]))
        sch.assume(["node-a"], p)
        sch.bind("node-a", p)
        victims.append(p["metadata"]["uid"])
    pre = client.create_pod(make_pod("pre", per_container=[{"pgpu": 1}]))
    r = extender.request("POST", "/scheduler/preemption", {
        "pod": pre,
        "nodeNameToMetaVictims": {
            "node-a": {"pods": [{"uid": u} for u in victims]}},
    })
    assert r.status_code == 200
    body = r.json()
    meta = body["nodeNameToMetaVictims"]
    assert len(meta["node-a"]["pods"]) == 1
    assert meta["node-a"]["pods"][0]["uid"] in victims
