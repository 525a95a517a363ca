"""Extender HTTP protocol tests: wire-compatible JSON in/out, plus the
hardening fixes (400 on malformed JSON — the reference panics,
routes.go:97-103)."""
from __future__ import annotations

import json

from tests.conftest import make_pod

GiB = 1024**3


def test_filter_partitions_nodes(cluster, extender):
    client, registry, app = cluster
    pod = client.create_pod(make_pod("p", core=30, memory=64 * GiB))
    r = extender.filter(pod, ["node-a", "node-b", "ghost"])
    assert r.status_code == 200
    body = r.json()
    assert sorted(body["nodenames"]) == ["node-a", "node-b"]
    assert "ghost" in body["failedNodes"]


def test_filter_requires_nodenames(cluster, extender):
    client, _, _ = cluster
    pod = client.create_pod(make_pod("p", core=30))
    r = extender.request("POST", "/scheduler/filter", {"pod": pod})
    assert r.status_code == 200
    assert "nodeCacheCapable" in r.json()["error"]


def test_filter_malformed_json_is_400_not_crash(extender):
    r = extender.request("POST", "/scheduler/filter", content=b"{nope")
    assert r.status_code == 400
    r2 = extender.request("POST", "/scheduler/priorities", content=b"[1,2")
    assert r2.status_code == 400


def test_priorities_returns_integer_scores(cluster, extender):
    client, _, _ = cluster
    pod = client.create_pod(make_pod("p", core=30, memory=64 * GiB))
    extender.filter(pod, ["node-a", "node-b"])
    r = extender.priorities(pod, ["node-a", "node-b"])
    assert r.status_code == 200
    out = r.json()
    assert {e["host"] for e in out} == {"node-a", "node-b"}
    for e in out:
        assert isinstance(e["score"], int)
        assert 0 <= e["score"] <= 10


def test_bind_happy_path(cluster, extender):
    client, _, _ = cluster
    pod = client.create_pod(make_pod("p", core=30, memory=64 * GiB))
    extender.filter(pod, ["node-a"])
    r = extender.bind(pod, "node-a")
    assert r.status_code == 200 and r.json() == {}
    bound = client.get_pod("default", "p")
    assert bound["spec"]["nodeName"] == "node-a"
    ann = bound["metadata"]["annotations"]
    assert ann["elasticgpu.io/assumed"] == "true"
    assert ann["elasticgpu.io/container-c0"] in {str(i) for i in range(8)}
    assert bound["metadata"]["labels"]["elasticgpu.io/assumed"] == "true"


def test_bind_uid_mismatch_rejected(cluster, extender):
    client, _, _ = cluster
    pod = client.create_pod(make_pod("p", core=30))
    stale = dict(pod)
    stale["metadata"] = dict(pod["metadata"], uid="other-uid")
    r = extender.bind(stale, "node-a")
    assert r.status_code == 500
    assert "UID" in r.json()["error"]


def test_bind_completed_pod_rejected(cluster, extender):
    client, _, _ = cluster
    pod = client.create_pod(make_pod("p", core=30))
    client.set_pod_phase("default", "p", "Succeeded")
    r = extender.bind(pod, "node-a")
    assert r.status_code == 500
    assert "completed" in r.json()["error"]


def test_bind_unknown_pod_rejected(cluster, extender):
    pod = make_pod("ghost", core=30)
    r = extender.bind(pod, "node-a")
    assert r.status_code == 500
    assert "not found" in r.json()["error"]


def test_bind_infeasible_rolls_back(cluster, extender):
    client, registry, _ = cluster
    big = client.create_pod(make_pod("big", per_container=[{"pgpu": 8}]))
    r = extender.bind(big, "node-a")
    assert r.status_code == 200
    too_big = client.create_pod(make_pod("more", per_container=[{"pgpu": 1}]))
    r = extender.bind(too_big, "node-a")
    assert r.status_code == 500
    # state unchanged: forget big, then 8 cards free again
    registry.default.forget_pod(client.get_pod("default", "big"))
    r = extender.bind(too_big, "node-a")
    assert r.status_code == 200


def test_non_gpu_pod_passes_through(cluster, extender):
    pod = {"metadata": {"name": "cpu", "namespace": "default", "uid": "u"},
           "spec": {"containers": [{"name": "c",
                                    "resources": {"requests": {"cpu": "1"}}}]}}
    r = extender.filter(pod, ["node-a", "node-b"])
    assert r.status_code == 200
    assert r.json()["nodenames"] == ["node-a", "node-b"]
    r = extender.priorities(pod, ["node-a"])
    assert r.json() == [{"host": "node-a", "score": 0}]


def test_status_version_metrics_healthz(cluster, extender):
    client, _, _ = cluster
    pod = client.create_pod(make_pod("p", core=30, memory=64 * GiB))
    extender.filter(pod, ["node-a"])
    extender.bind(pod, "node-a")
    r = extender.request("GET", "/scheduler/status")
    st = r.json()
    assert "node-a" in st["gpushare"]["nodes"]
    gpus = st["gpushare"]["nodes"]["node-a"]["gpus"]
    assert len(gpus) == 8
    assert any(g["core_available"] == 70 for g in gpus)
    pods = st["gpushare"]["nodes"]["node-a"]["pods"]
    assert list(pods.values()) == [[[0]]] or \
        any(len(v) == 1 and len(v[0]) == 1 for v in pods.values())

    assert extender.request("GET", "/version").json()["target"].startswith("MI355X")
    m = extender.request("GET", "/metrics")
    assert b"egs_pods_scheduled_total" in m.content
    assert extender.request("GET", "/healthz").json() == {"ok": True}
    stacks = extender.request("GET", "/debug/stacks")
    assert b"thread" in stacks.content
    assert extender.request("GET", "/nope").status_code == 404


def test_wire_format_matches_extender_v1(cluster, extender):
    """Exact JSON field names from k8s.io/kube-scheduler/extender/v1."""
    client, _, _ = cluster
    pod = client.create_pod(make_pod("p", core=30))
    raw = json.dumps({"pod": pod, "nodenames": ["node-a"],
                      "nodes": None}).encode()
    r = extender.request("POST", "/scheduler/filter", content=raw)
    body = r.json()
    assert set(body.keys()) <= {"nodenames", "failedNodes", "nodes",
                                "failedAndUnresolvable", "error"}


def test_spread_containers_end_to_end(cluster, extender):
    """A 3-container pod with elasticgpu.io/spread-containers=true lands on
    three distinct cards, through the full HTTP pipeline."""
    client, registry, _ = cluster
    pod = make_pod("sp", containers=3, core=20, memory=GiB)
    pod["metadata"]["annotations"] = {"elasticgpu.io/spread-containers": "true"}
    created = client.create_pod(pod)
    r = extender.filter(created, ["node-a"])
    assert r.json()["nodenames"] == ["node-a"]
    assert extender.bind(created, "node-a").status_code == 200
    bound = client.get_pod("default", "sp")
    from elastic_gpu_scheduler_amd.k8s import objects as obj
    alloc = obj.parse_allocation(bound)
    cards = [a[0] for a in alloc]
    assert len(set(cards)) == 3, cards


def test_debug_profile_collapsed_stacks(cluster, extender):
    r = extender.request("GET", "/debug/profile")
    assert r.status_code == 200
    text = r.text
    # every line is "frame;frame;... count"
    line = text.strip().splitlines()[0]
    stack, _, count = line.rpartition(" ")
    assert int(count) >= 1
    assert ":" in stack


def test_non_utf8_body_is_400(extender):
    r = extender.request("POST", "/scheduler/filter", content=b"\xe1\x97\x9f")
    assert r.status_code == 400
    r = extender.request("POST", "/scheduler/bind", content=b"\xff\xfe")
    assert r.status_code == 400


def test_qgpu_mode_end_to_end(fake_client):
    """qgpu mode: only qgpu resources route; full cycle over the app."""
    from elastic_gpu_scheduler_amd.scheduler.service import SchedulerRegistry
    from elastic_gpu_scheduler_amd.server.app import make_app
    from tests.conftest import ExtenderClient, make_node

    fake_client.add_node(make_node("node-a"))
    registry = SchedulerRegistry(fake_client, mode="qgpu")
    ext = ExtenderClient(make_app(registry))

    qpod = fake_client.create_pod({
        "metadata": {"name": "q", "namespace": "default", "uid": "qu"},
        "spec": {"containers": [{"name": "c", "resources": {"requests": {
            "elasticgpu.io/qgpu-core": "40",
            "elasticgpu.io/qgpu-memory": "32Gi"}}}]},
        "status": {"phase": "Pending"}})
    r = ext.filter(qpod, ["node-a"])
    assert r.json()["nodenames"] == ["node-a"]
    assert ext.bind(qpod, "node-a").status_code == 200
    bound = fake_client.get_pod("default", "q")
    assert bound["spec"]["nodeName"] == "node-a"
    assert bound["metadata"]["annotations"]["elasticgpu.io/container-c"] == "0"

    # gpushare pods are NOT managed in qgpu mode: pass-through
    gpod = fake_client.create_pod(make_pod("g", core=30))
    r = ext.filter(gpod, ["node-a"])
    assert r.json()["nodenames"] == ["node-a"]  # untouched


def test_pgpu_mode_end_to_end(fake_client):
    from elastic_gpu_scheduler_amd.scheduler.service import SchedulerRegistry
    from elastic_gpu_scheduler_amd.server.app import make_app
    from tests.conftest import ExtenderClient, make_node

    fake_client.add_node(make_node("node-a", cards=2))
    registry = SchedulerRegistry(fake_client, mode="pgpu")
    ext = ExtenderClient(make_app(registry))
    pod = fake_client.create_pod(make_pod("p", per_container=[{"pgpu": 2}]))
    r = ext.filter(pod, ["node-a"])
    assert r.json()["nodenames"] == ["node-a"]
    assert ext.bind(pod, "node-a").status_code == 200
    bound = fake_client.get_pod("default", "p")
    assert bound["metadata"]["annotations"]["elasticgpu.io/container-c0"] == "0,1"


def test_hist_quantile_math():
    """log2-us histogram quantile helper used by /debug/latency."""
    from elastic_gpu_scheduler_amd.server.app import _hist_quantile

    h = {"count": 0, "buckets": [(1, 0), (2, 0)]}
    assert _hist_quantile(h, 0.5) is None
    # 10 samples <=2us, 90 samples <=1024us
    h = {"count": 100, "buckets": [(2, 10), (1024, 90)]}
    assert _hist_quantile(h, 0.05) == 2
    assert _hist_quantile(h, 0.5) == 1024
    assert _hist_quantile(h, 0.99) == 1024
