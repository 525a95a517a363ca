"""Scheduler service: warm-start recovery, lazy node fill, bind retry and
rollback, tombstones — the crash-consistency behaviors SURVEY.md §5 calls
out."""
from __future__ import annotations

import pytest

from elastic_gpu_scheduler_amd.k8s import objects as obj
from elastic_gpu_scheduler_amd.k8s.client import ConflictError, FakeKubeClient
from elastic_gpu_scheduler_amd.scheduler.service import (BindError,
                                                         GPUUnitScheduler,
                                                         SchedulerRegistry)
from tests.conftest import make_node, make_pod

GiB = 1024**3


def test_warm_start_rebuilds_from_annotations():
    """Annotations on bound pods ARE the checkpoint (reference
    scheduler.go:86-106): a fresh scheduler must rebuild accounting."""
    client = FakeKubeClient()
    client.add_node(make_node("n1"))
    sch1 = GPUUnitScheduler(client)
    pod = client.create_pod(make_pod("p", core=40, memory=100 * GiB))
    sch1.assume(["n1"], pod)
    sch1.bind("n1", client.get_pod("default", "p"))

    # process restart:
    sch2 = GPUUnitScheduler(client)
    sch2._ensure_node("n1")
    devs = sch2.state.node_devices("n1")
    assert any(d.core_avail == 60 for d in devs)
    assert sch2.state.known_pod(obj.pod_uid(pod))


def test_crash_between_annotate_and_bind_is_recoverable():
    """If the process dies after the annotation Update but before the
    Binding POST, the pod has no nodeName — our scheduled-node annotation
    still attributes it, and re-binding is idempotent."""
    client = FakeKubeClient()
    client.add_node(make_node("n1"))
    sch1 = GPUUnitScheduler(client)
    pod = client.create_pod(make_pod("p", core=40, memory=100 * GiB))
    sch1.assume(["n1"], pod)
    option = sch1.state.allocate("n1", obj.pod_uid(pod),
                                 obj.pod_gpu_request(pod))
    # simulate: annotation written, bind POST never happened
    annotated = obj.apply_allocation(pod, [list(a) for a in option.allocated],
                                     node="n1", score=option.score)
    client.update_pod(annotated)

    sch2 = GPUUnitScheduler(client)  # warm start picks it up via scheduled-node
    sch2._ensure_node("n1")
    assert sch2.state.known_pod(obj.pod_uid(pod))
    assert any(d.core_avail == 60 for d in sch2.state.node_devices("n1"))
    # kube-scheduler retries the bind: must succeed without double-charging
    sch2.bind("n1", client.get_pod("default", "p"))
    assert client.get_pod("default", "p")["spec"]["nodeName"] == "n1"
    assert any(d.core_avail == 60 for d in sch2.state.node_devices("n1"))


def test_lazy_node_fill_replays_existing_pods():
    client = FakeKubeClient()
    client.add_node(make_node("n1"))
    sch1 = GPUUnitScheduler(client)
    pod = client.create_pod(make_pod("p", core=25, memory=10 * GiB))
    sch1.assume(["n1"], pod)
    sch1.bind("n1", client.get_pod("default", "p"))

    sch2 = GPUUnitScheduler(client)
    sch2.invalidate_node("n1")  # force a cold cache for the lazy-fill path
    ok, failed = sch2.assume(["n1"], client.create_pod(make_pod("q", core=80)))
    assert ok == ["n1"]
    # _ensure_node replayed the assumed pod: 25 core already used on one card
    assert any(d.core_avail == 75 for d in sch2.state.node_devices("n1"))


def test_bind_conflict_retries_with_fresh_pod():
    client = FakeKubeClient()
    client.add_node(make_node("n1"))
    sch = GPUUnitScheduler(client)
    pod = client.create_pod(make_pod("p", core=30, memory=GiB))
    sch.assume(["n1"], pod)
    # out-of-band update bumps resourceVersion -> first update conflicts
    fresh = client.get_pod("default", "p")
    fresh["metadata"].setdefault("labels", {})["x"] = "y"
    client.update_pod(fresh)
    sch.bind("n1", pod)  # stale resourceVersion in hand; must retry internally
    bound = client.get_pod("default", "p")
    assert bound["spec"]["nodeName"] == "n1"
    assert bound["metadata"]["labels"]["x"] == "y"  # retry used the fresh pod


def test_bind_failure_rolls_back_allocation():
    class FailingClient(FakeKubeClient):
        def bind_pod(self, namespace, name, node):
            raise RuntimeError("apiserver down")

    client = FailingClient()
    client.add_node(make_node("n1", cards=1))
    sch = GPUUnitScheduler(client)
    pod = client.create_pod(make_pod("p", core=30, memory=GiB))
    sch.assume(["n1"], pod)
    with pytest.raises(RuntimeError):
        sch.bind("n1", pod)
    # allocation rolled back: card fully free
    d = sch.state.node_devices("n1")[0]
    assert d.core_avail == 100 and d.mem_avail == 288 * GiB


def test_released_tombstone_blocks_readd():
    """A DELETE seen before the last MODIFIED must not resurrect accounting
    (reference releasedPodMap, scheduler.go:47, 247-281)."""
    client = FakeKubeClient()
    client.add_node(make_node("n1"))
    sch = GPUUnitScheduler(client)
    pod = client.create_pod(make_pod("p", core=30, memory=GiB))
    sch.assume(["n1"], pod)
    sch.bind("n1", pod)
    bound = client.get_pod("default", "p")
    sch.forget_pod(bound)
    assert sch.released_pod(bound)
    sch.add_pod(bound)  # late MODIFIED replay
    assert not sch.state.known_pod(obj.pod_uid(bound))
    d0 = sch.state.node_devices("n1")
    assert all(d.core_avail == 100 for d in d0)


def test_registry_routes_by_resource_name():
    client = FakeKubeClient()
    reg = SchedulerRegistry(client)
    gpu_pod = make_pod("p", core=10)
    assert reg.for_pod(gpu_pod) is reg.default
    cpu_pod = {"spec": {"containers": [{"resources": {"requests": {"cpu": "1"}}}]}}
    assert reg.for_pod(cpu_pod) is None
    qgpu_pod = {"spec": {"containers": [{"resources": {"requests": {
        "elasticgpu.io/qgpu-core": "50"}}}]}}
    assert reg.for_pod(qgpu_pod) is reg.default


def test_qgpu_mode_registry():
    client = FakeKubeClient()
    reg = SchedulerRegistry(client, mode="qgpu")
    qgpu_pod = {"spec": {"containers": [{"resources": {"requests": {
        "elasticgpu.io/qgpu-core": "50"}}}]}}
    gpushare_pod = make_pod("p", core=10)
    assert reg.for_pod(qgpu_pod) is reg.default
    assert reg.for_pod(gpushare_pod) is None


def test_pgpu_whole_card_semantics():
    client = FakeKubeClient()
    client.add_node(make_node("n1", cards=2))
    sch = GPUUnitScheduler(client)
    pod = client.create_pod(make_pod("p", per_container=[{"pgpu": 2}]))
    ok, _ = sch.assume(["n1"], pod)
    assert ok == ["n1"]
    sch.bind("n1", client.get_pod("default", "p"))
    devs = sch.state.node_devices("n1")
    assert all(d.core_avail == 0 for d in devs)
