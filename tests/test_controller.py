"""Reconcile controller: watch events keep accounting honest (reference
pkg/controller/controller.go syncPod semantics)."""
from __future__ import annotations

import time

from elastic_gpu_scheduler_amd.controller.controller import Controller
from elastic_gpu_scheduler_amd.k8s import objects as obj
from elastic_gpu_scheduler_amd.k8s.client import FakeKubeClient
from elastic_gpu_scheduler_amd.scheduler.service import SchedulerRegistry
from tests.conftest import make_node, make_pod

GiB = 1024**3


def wait_until(pred, timeout=5.0):
    deadline = time.time() + timeout
    while time.time() < deadline:
        try:
            if pred():
                return True
        except RuntimeError:
            pass  # e.g. node cache not yet filled by the controller
        time.sleep(0.01)
    return False


def make_stack():
    client = FakeKubeClient()
    client.add_node(make_node("n1"))
    registry = SchedulerRegistry(client)
    ctrl = Controller(client, registry, workers=2, resync_seconds=3600)
    ctrl.start()
    return client, registry, ctrl


def test_completed_pod_is_released():
    client, registry, ctrl = make_stack()
    try:
        sch = registry.default
        pod = client.create_pod(make_pod("p", core=40, memory=GiB))
        sch.assume(["n1"], pod)
        sch.bind("n1", pod)
        assert any(d.core_avail == 60 for d in sch.state.node_devices("n1"))
        client.set_pod_phase("default", "p", "Succeeded")
        assert wait_until(lambda: all(
            d.core_avail == 100 for d in sch.state.node_devices("n1")))
    finally:
        ctrl.stop()


def test_deleted_pod_is_released():
    client, registry, ctrl = make_stack()
    try:
        sch = registry.default
        pod = client.create_pod(make_pod("p", core=40, memory=GiB))
        sch.assume(["n1"], pod)
        sch.bind("n1", pod)
        client.delete_pod("default", "p")
        assert wait_until(lambda: all(
            d.core_avail == 100 for d in sch.state.node_devices("n1")))
        # late MODIFIED must not resurrect (tombstone)
        sch.add_pod(client_pod_with_node(pod))
        assert all(d.core_avail == 100 for d in sch.state.node_devices("n1"))
    finally:
        ctrl.stop()


def client_pod_with_node(pod):
    p = dict(pod)
    p["spec"] = dict(pod["spec"], nodeName="n1")
    return p


def test_externally_assigned_pod_is_accounted():
    """A pod bound by another scheduler replica / before our startup shows
    up via the watch and must be charged (reference assignPod path,
    controller.go:174-180)."""
    client, registry, ctrl = make_stack()
    try:
        sch = registry.default
        pod = make_pod("ext", core=30, memory=GiB)
        pod["metadata"]["labels"] = {"elasticgpu.io/assumed": "true"}
        pod["metadata"]["annotations"] = {
            "elasticgpu.io/assumed": "true",
            "elasticgpu.io/container-c0": "0",
        }
        pod["spec"]["nodeName"] = "n1"
        client.create_pod(pod)
        assert wait_until(lambda: any(
            d.core_avail == 70 for d in sch.state.node_devices("n1")))
    finally:
        ctrl.stop()


def test_resync_evicts_vanished_pods():
    client, registry, ctrl = make_stack()
    try:
        sch = registry.default
        pod = client.create_pod(make_pod("p", core=40, memory=GiB))
        sch.assume(["n1"], pod)
        sch.bind("n1", pod)
        assert ctrl.wait_idle()  # drain the bind's own MODIFIED events first
        # vanish without a DELETE event: remove directly from the store
        with client._mu:
            client._pods.clear()
        # Eviction needs TWO consecutive relist misses (a single-relist
        # evict races pods bound between list_pods() and the sweep).
        ctrl.resync_once()
        assert any(d.core_avail == 60 for d in sch.state.node_devices("n1"))
        ctrl.resync_once()
        assert all(d.core_avail == 100 for d in sch.state.node_devices("n1"))
    finally:
        ctrl.stop()


def test_resync_snapshot_race_does_not_evict_fresh_pod():
    """a pod bound between list_pods() (the snapshot) and the
    sweep must NOT be forgotten — one stale relist marks it missing, and a
    later relist that sees it clears the mark."""
    client, registry, ctrl = make_stack()
    try:
        sch = registry.default
        pod = client.create_pod(make_pod("p", core=40, memory=GiB))
        sch.assume(["n1"], pod)
        sch.bind("n1", pod)
        assert ctrl.wait_idle()
        # Simulate a list snapshot taken BEFORE this pod was committed.
        real_list = client.list_pods
        client.list_pods = lambda *a, **k: []
        ctrl.resync_once()
        # Still accounted after one stale relist.
        assert any(d.core_avail == 60 for d in sch.state.node_devices("n1"))
        # The next relist sees the pod again: the missing mark is cleared...
        client.list_pods = real_list
        ctrl.resync_once()
        assert any(d.core_avail == 60 for d in sch.state.node_devices("n1"))
        # ...so even another stale relist does not evict it.
        client.list_pods = lambda *a, **k: []
        ctrl.resync_once()
        assert any(d.core_avail == 60 for d in sch.state.node_devices("n1"))
    finally:
        ctrl.stop()


def test_non_gpu_pods_ignored():
    client, registry, ctrl = make_stack()
    try:
        client.create_pod({
            "metadata": {"name": "cpu", "namespace": "default"},
            "spec": {"containers": [{"name": "c",
                                     "resources": {"requests": {"cpu": "1"}}}],
                     "nodeName": "n1"},
            "status": {"phase": "Running"},
        })
        time.sleep(0.1)
        # never even cached the node: non-GPU pods are filtered at the watch
        assert not registry.default.state.has_node("n1")
    finally:
        ctrl.stop()


def test_sync_failure_retries_with_backoff():
    """A transient sync failure requeues the pod instead of dropping it."""
    client, registry, ctrl = make_stack()
    try:
        sch = registry.default
        calls = {"n": 0}
        orig = sch.add_pod

        def flaky_add(pod):
            calls["n"] += 1
            if calls["n"] < 3:
                raise RuntimeError("transient")
            orig(pod)

        sch.add_pod = flaky_add
        pod = make_pod("ext", core=30, memory=GiB)
        pod["metadata"]["labels"] = {"elasticgpu.io/assumed": "true"}
        pod["metadata"]["annotations"] = {"elasticgpu.io/assumed": "true",
                                          "elasticgpu.io/container-c0": "0"}
        pod["spec"]["nodeName"] = "n1"
        client.create_pod(pod)
        assert wait_until(lambda: any(
            d.core_avail == 70 for d in sch.state.node_devices("n1")))
        assert calls["n"] >= 3
    finally:
        ctrl.stop()


def test_resync_refreshes_changed_node_inventory():
    """When the agent republishes a node's inventory (resourceVersion
    bumps), the resync evicts the cached allocator; the next use re-reads
    the fresh inventory and replays assumed pods."""
    client, registry, ctrl = make_stack()
    try:
        sch = registry.default
        pod = client.create_pod(make_pod("p", core=25, memory=GiB))
        sch.assume(["n1"], pod)
        sch.bind("n1", pod)
        ctrl.resync_once()  # records the baseline resourceVersion
        assert len(sch.state.node_devices("n1")) == 8

        # agent publishes a shrunk inventory (e.g. a sick card excluded)
        import json as _json

        cards = [{"index": i, "memory_bytes": 288 * GiB, "core": 100}
                 for i in range(4)]
        client.patch_node_annotations("n1", {
            "elasticgpu.io/gpu-inventory": _json.dumps({"cards": cards})})
        ctrl.resync_once()  # detects the change and evicts
        ok, _ = sch.assume(["n1"], client.create_pod(make_pod("q", core=10)))
        assert ok == ["n1"]
        devs = sch.state.node_devices("n1")
        assert len(devs) == 4  # fresh inventory picked up
        # the bound pod's accounting survived the refresh (replayed)
        assert any(d.core_avail == 75 for d in devs)
    finally:
        ctrl.stop()


def test_node_watch_invalidates_cache_immediately():
    """Agent republish (node annotation patch) must refresh the scheduler's
    node cache via the NODE WATCH — not only at the next periodic resync.
    (The reference creates a node informer and never consults it,
    controller.go:97-99.)"""
    import json as _json

    client, registry, ctrl = make_stack()
    try:
        sch = registry.default
        pod = client.create_pod(make_pod("p", core=40, memory=GiB))
        sch.assume(["n1"], pod)
        sch.bind("n1", pod)
        assert ctrl.wait_idle()
        assert sch.state.has_node("n1")
        # the agent publishes a richer inventory: 2 cards instead of default
        inv = {"cards": [{"index": 0, "core": 100,
                          "memory_bytes": 288 * GiB},
                         {"index": 1, "core": 100,
                          "memory_bytes": 288 * GiB}]}
        client.patch_node_annotations(
            "n1", {"elasticgpu.io/gpu-inventory": _json.dumps(inv)})
        # invalidation is synchronous with the fake's watch fan-out: the
        # cache entry is gone; next use refills from the new annotation
        assert wait_until(lambda: not sch.state.has_node("n1"))
        ok, _ = sch.assume(["n1"], client.create_pod(
            make_pod("p2", core=40, memory=GiB)))
        assert ok == ["n1"]
        devs = sch.state.node_devices("n1")
        assert len(devs) == 2
        # the bound pod was replayed into the refreshed cache
        assert any(d.core_avail == 60 for d in devs)
    finally:
        ctrl.stop()


def test_node_watch_delete_evicts():
    client, registry, ctrl = make_stack()
    try:
        sch = registry.default
        pod = client.create_pod(make_pod("p", core=40, memory=GiB))
        sch.assume(["n1"], pod)
        assert sch.state.has_node("n1")
        client.delete_node("n1")
        assert wait_until(lambda: not sch.state.has_node("n1"))
    finally:
        ctrl.stop()
