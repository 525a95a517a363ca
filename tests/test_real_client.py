"""RealKubeClient against an httpx MockTransport standing in for the
apiserver REST surface (no live cluster in CI)."""
from __future__ import annotations

import json

import httpx
import pytest

from elastic_gpu_scheduler_amd.k8s.client import (ConflictError, NotFoundError,
                                                  RealKubeClient)


class MockApiserver:
    """Tiny stateful apiserver over httpx.MockTransport."""

    def __init__(self):
        self.pods = {}
        self.bindings = []
        self.events = []
        self.patches = []

    def handler(self, request: httpx.Request) -> httpx.Response:
        path = request.url.path
        method = request.method
        if method == "GET" and path.startswith("/api/v1/namespaces/"):
            parts = path.split("/")
            key = f"{parts[4]}/{parts[6]}"
            if key not in self.pods:
                return httpx.Response(404, json={"message": "not found"})
            return httpx.Response(200, json=self.pods[key])
        if method == "GET" and path == "/api/v1/pods":
            items = list(self.pods.values())
            sel = request.url.params.get("labelSelector", "")
            if sel:
                k, _, v = sel.partition("=")
                items = [p for p in items
                         if (p.get("metadata", {}).get("labels", {})
                             or {}).get(k) == v]
            return httpx.Response(200, json={"items": items})
        if method == "PUT" and "/pods/" in path:
            pod = json.loads(request.content)
            parts = path.split("/")
            key = f"{parts[4]}/{parts[6]}"
            if key not in self.pods:
                return httpx.Response(404, json={})
            if pod["metadata"].get("resourceVersion") != \
                    self.pods[key]["metadata"].get("resourceVersion"):
                return httpx.Response(409, json={"message": "conflict"})
            pod["metadata"]["resourceVersion"] = str(
                int(pod["metadata"]["resourceVersion"]) + 1)
            self.pods[key] = pod
            return httpx.Response(200, json=pod)
        if method == "POST" and path.endswith("/binding"):
            self.bindings.append(json.loads(request.content))
            return httpx.Response(201, json={})
        if method == "POST" and path.endswith("/events"):
            self.events.append(json.loads(request.content))
            return httpx.Response(201, json={})
        if method == "GET" and path.startswith("/api/v1/nodes/"):
            name = path.rsplit("/", 1)[1]
            return httpx.Response(200, json={"metadata": {"name": name}})
        if method == "GET" and path == "/api/v1/nodes":
            return httpx.Response(200, json={"items": []})
        if method == "PATCH" and path.startswith("/api/v1/nodes/"):
            self.patches.append((path, json.loads(request.content)))
            return httpx.Response(200, json={"metadata": {}})
        return httpx.Response(404, json={"message": f"no route {path}"})


@pytest.fixture
def api():
    mock = MockApiserver()
    client = RealKubeClient("https://apiserver", token="tok",
                            transport=httpx.MockTransport(mock.handler))
    return mock, client


def test_get_update_conflict_flow(api):
    mock, client = api
    mock.pods["default/p"] = {"metadata": {"name": "p", "namespace": "default",
                                           "uid": "u", "resourceVersion": "1"}}
    pod = client.get_pod("default", "p")
    assert pod["metadata"]["uid"] == "u"
    pod.setdefault("metadata", {}).setdefault("annotations", {})["k"] = "v"
    updated = client.update_pod(pod)
    assert updated["metadata"]["resourceVersion"] == "2"
    # stale update -> typed ConflictError (not string matching)
    with pytest.raises(ConflictError):
        client.update_pod(pod)
    with pytest.raises(NotFoundError):
        client.get_pod("default", "ghost")


def test_bind_posts_binding_subresource(api):
    mock, client = api
    mock.pods["default/p"] = {"metadata": {"name": "p", "namespace": "default"}}
    client.bind_pod("default", "p", "node-9")
    assert mock.bindings[0]["target"]["name"] == "node-9"
    assert mock.bindings[0]["kind"] == "Binding"


def test_list_pods_label_selector(api):
    mock, client = api
    mock.pods["default/a"] = {"metadata": {"name": "a", "namespace": "default",
                                           "labels": {"elasticgpu.io/assumed": "true"}}}
    mock.pods["default/b"] = {"metadata": {"name": "b", "namespace": "default"}}
    out = client.list_pods(label_selector={"elasticgpu.io/assumed": "true"})
    assert [p["metadata"]["name"] for p in out] == ["a"]


def test_patch_node_and_events(api):
    mock, client = api
    client.patch_node_annotations("n1", {"a": "b"})
    assert mock.patches[0][0] == "/api/v1/nodes/n1"
    assert mock.patches[0][1] == {"metadata": {"annotations": {"a": "b"}}}
    client.create_event("default", {"reason": "Scheduled"})
    assert mock.events[0]["reason"] == "Scheduled"


def test_scheduler_service_over_rest_client(api):
    """Full verb cycle through RealKubeClient (REST) instead of the fake:
    the interface seam the reference lacks."""
    from elastic_gpu_scheduler_amd.scheduler.service import GPUUnitScheduler

    mock, client = api
    # apiserver knows one 8-card MI355X node
    GiB = 1024**3

    real_get_node = mock.handler

    def handler(request):
        if request.method == "GET" and request.url.path == "/api/v1/nodes/gpu-1":
            return httpx.Response(200, json={
                "metadata": {"name": "gpu-1"},
                "status": {"allocatable": {
                    "elasticgpu.io/gpu-core": "800",
                    "elasticgpu.io/gpu-memory": str(8 * 288 * GiB)}}})
        return real_get_node(request)

    client._client._transport = httpx.MockTransport(handler)

    mock.pods["default/p"] = {
        "metadata": {"name": "p", "namespace": "default", "uid": "u-1",
                     "resourceVersion": "1"},
        "spec": {"containers": [{"name": "c", "resources": {"requests": {
            "elasticgpu.io/gpu-core": "30",
            "elasticgpu.io/gpu-memory": str(48 * GiB)}}}]},
        "status": {"phase": "Pending"},
    }
    sch = GPUUnitScheduler(client)
    pod = client.get_pod("default", "p")
    ok, failed = sch.assume(["gpu-1", "ghost"], pod)
    assert ok == ["gpu-1"], failed
    scores = sch.score(["gpu-1"], pod)
    assert 0 <= scores[0] <= 10
    sch.bind("gpu-1", pod)
    # the annotation Update + Binding both hit the REST surface
    assert mock.bindings[0]["target"]["name"] == "gpu-1"
    stored = mock.pods["default/p"]
    assert stored["metadata"]["annotations"]["elasticgpu.io/container-c"] \
        in {str(i) for i in range(8)}
    sch.flush_events()
    assert mock.events, "bind must emit a scheduling event"


def test_watch_pods_streams_and_reconnects():
    """RealKubeClient.watch_pods: parses the apiserver's JSON-lines watch
    stream, delivers events, and survives a dropped stream (reconnect)."""
    import threading
    import time

    events_delivered = []
    connections = {"n": 0}

    def handler(request: httpx.Request) -> httpx.Response:
        if request.url.params.get("watch") == "true":
            connections["n"] += 1
            if connections["n"] == 1:
                body = (json.dumps({"type": "ADDED",
                                    "object": {"metadata": {"name": "w1"}}})
                        + "\n" +
                        json.dumps({"type": "MODIFIED",
                                    "object": {"metadata": {"name": "w1"}}})
                        + "\n")
            else:
                body = json.dumps({"type": "DELETED",
                                   "object": {"metadata": {"name": "w1"}}}) + "\n"
            return httpx.Response(200, content=body.encode())
        return httpx.Response(404)

    client = RealKubeClient("https://apiserver", token="t",
                            transport=httpx.MockTransport(handler))
    # shrink the reconnect backoff for the test
    got = threading.Event()

    def on_event(etype, obj):
        events_delivered.append((etype, obj.get("metadata", {}).get("name")))
        if etype == "DELETED":
            got.set()

    unsubscribe = client.watch_pods(on_event)
    try:
        assert got.wait(timeout=10), events_delivered
    finally:
        unsubscribe()
    assert ("ADDED", "w1") in events_delivered
    assert ("MODIFIED", "w1") in events_delivered
    assert ("DELETED", "w1") in events_delivered  # arrived via reconnect
    assert connections["n"] >= 2
