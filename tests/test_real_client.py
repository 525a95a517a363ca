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
                                    "object": {"metadata": {"name": "w1",
                                               "resourceVersion": "11"}}})
                        + "\n" +
                        json.dumps({"type": "MODIFIED",
                                    "object": {"metadata": {"name": "w1",
                                               "resourceVersion": "12"}}})
                        + "\n")
            else:
                body = json.dumps({"type": "DELETED",
                                   "object": {"metadata": {"name": "w1",
                                              "resourceVersion": "13"}}}) + "\n"
            return httpx.Response(200, content=body.encode())
        # initial relist establishing the snapshot + RV to watch from
        return httpx.Response(200, json={
            "kind": "PodList", "metadata": {"resourceVersion": "10"},
            "items": []})

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


def test_watch_resumes_from_resource_version():
    """a dropped watch RESUMES from the last seen
    resourceVersion — the event created during the gap is delivered by the
    resumed watch, and no second full relist happens."""
    import threading

    lists = {"n": 0}
    watch_rvs = []
    delivered = []
    done = threading.Event()

    def handler(request: httpx.Request) -> httpx.Response:
        if request.url.params.get("watch") == "true":
            watch_rvs.append(request.url.params.get("resourceVersion"))
            if len(watch_rvs) == 1:
                # one event, then the stream drops
                body = json.dumps({"type": "ADDED",
                                   "object": {"metadata": {"name": "a",
                                              "resourceVersion": "21"}}}) + "\n"
            else:
                # resumed watch delivers what happened during the gap
                body = json.dumps({"type": "ADDED",
                                   "object": {"metadata": {"name": "gap-pod",
                                              "resourceVersion": "22"}}}) + "\n"
            return httpx.Response(200, content=body.encode())
        lists["n"] += 1
        return httpx.Response(200, json={
            "kind": "PodList", "metadata": {"resourceVersion": "20"},
            "items": [{"metadata": {"name": "seed", "resourceVersion": "19"}}]})

    client = RealKubeClient("https://apiserver", token="t",
                            transport=httpx.MockTransport(handler))

    def on_event(etype, obj):
        name = obj.get("metadata", {}).get("name")
        delivered.append((etype, name))
        if name == "gap-pod":
            done.set()

    unsubscribe = client.watch_pods(on_event)
    try:
        assert done.wait(timeout=10), delivered
    finally:
        unsubscribe()
    # initial snapshot delivered as synthetic MODIFIED
    assert ("MODIFIED", "seed") in delivered
    assert ("ADDED", "gap-pod") in delivered  # nothing lost in the gap
    assert lists["n"] == 1, "reconnect must NOT trigger a relist storm"
    assert watch_rvs[0] == "20"   # watch starts from the list's RV
    assert watch_rvs[1] == "21"   # resume from the last DELIVERED event's RV


def test_watch_handles_410_gone_with_single_relist():
    """An ERROR event with Status code 410 (RV fell out of etcd's window)
    triggers exactly one relist, then watching resumes from the fresh RV."""
    import threading

    lists = {"n": 0}
    watch_calls = {"n": 0}
    delivered = []
    done = threading.Event()

    def handler(request: httpx.Request) -> httpx.Response:
        if request.url.params.get("watch") == "true":
            watch_calls["n"] += 1
            if watch_calls["n"] == 1:
                body = json.dumps({
                    "type": "ERROR",
                    "object": {"kind": "Status", "code": 410,
                               "reason": "Expired"}}) + "\n"
            else:
                body = json.dumps({"type": "ADDED",
                                   "object": {"metadata": {"name": "fresh",
                                              "resourceVersion": "31"}}}) + "\n"
            return httpx.Response(200, content=body.encode())
        lists["n"] += 1
        return httpx.Response(200, json={
            "kind": "PodList",
            "metadata": {"resourceVersion": str(30 + lists["n"])},
            "items": []})

    client = RealKubeClient("https://apiserver", token="t",
                            transport=httpx.MockTransport(handler))

    def on_event(etype, obj):
        delivered.append((etype, obj.get("metadata", {}).get("name")))
        if obj.get("metadata", {}).get("name") == "fresh":
            done.set()

    unsubscribe = client.watch_pods(on_event)
    try:
        assert done.wait(timeout=10), delivered
    finally:
        unsubscribe()
    assert lists["n"] == 2  # initial + exactly one post-410 relist
    assert not any(n == "Status" for _, n in delivered)  # ERROR not delivered


def test_from_kubeconfig_client_certificates(tmp_path):
    """A kind/minikube-style kubeconfig (inline base64 client cert/key +
    CA data) must configure mTLS — r1 read only user.token and could not
    connect to a default kind cluster at all. Real PKI
    material: the client loads it into an SSLContext at construction."""
    import base64

    from elastic_gpu_scheduler_amd.testing import generate_pki

    pki = generate_pki(tmp_path)
    fake_ca = open(pki["ca_crt"], "rb").read()
    fake_crt = open(pki["client_crt"], "rb").read()
    fake_key = open(pki["client_key"], "rb").read()
    cfg = {
        "current-context": "kind-kind",
        "contexts": [{"name": "kind-kind",
                      "context": {"cluster": "kind", "user": "kind-user"}}],
        "clusters": [{"name": "kind", "cluster": {
            "server": "https://127.0.0.1:6443",
            "certificate-authority-data":
                base64.b64encode(fake_ca).decode()}}],
        "users": [{"name": "kind-user", "user": {
            "client-certificate-data": base64.b64encode(fake_crt).decode(),
            "client-key-data": base64.b64encode(fake_key).decode()}}],
    }
    captured = {}

    import elastic_gpu_scheduler_amd.k8s.client as client_mod

    class CapturingClient:
        def __init__(self, **kw):
            captured.update(kw)

        def close(self):
            pass

    real_httpx_client = httpx.Client
    httpx.Client = lambda **kw: CapturingClient(**kw)
    try:
        c = client_mod.RealKubeClient.from_kubeconfig(cfg)
    finally:
        httpx.Client = real_httpx_client
    import os
    import ssl

    try:
        assert captured["base_url"] == "https://127.0.0.1:6443"
        # httpx 0.28 drops cert=(crt, key) when verify is a CA path, so the
        # client must hand httpx a ready SSLContext with the cert loaded.
        assert isinstance(captured["verify"], ssl.SSLContext)
        tmp = c._tmpdir.name
        with open(os.path.join(tmp, "client.crt"), "rb") as f:
            assert f.read() == fake_crt
        with open(os.path.join(tmp, "client.key"), "rb") as f:
            assert f.read() == fake_key
        with open(os.path.join(tmp, "ca.crt"), "rb") as f:
            assert f.read() == fake_ca
        assert "Authorization" not in captured["headers"]
    finally:
        c.close()
    # temp material is wiped on close
    assert not os.path.exists(tmp)


def test_from_kubeconfig_token_and_ca_paths(tmp_path):
    ca = tmp_path / "ca.crt"
    ca.write_text("ca")
    tok = tmp_path / "token"
    tok.write_text("sekrit\n")
    cfg = {
        "current-context": "c",
        "contexts": [{"name": "c", "context": {"cluster": "cl", "user": "u"}}],
        "clusters": [{"name": "cl", "cluster": {
            "server": "https://h:6443", "certificate-authority": "ca.crt"}}],
        "users": [{"name": "u", "user": {"tokenFile": str(tok)}}],
    }
    captured = {}

    class CapturingClient:
        def __init__(self, **kw):
            captured.update(kw)

        def close(self):
            pass

    real_httpx_client = httpx.Client
    httpx.Client = lambda **kw: CapturingClient(**kw)
    try:
        c = RealKubeClient.from_kubeconfig(cfg, base_dir=tmp_path)
    finally:
        httpx.Client = real_httpx_client
    assert captured["headers"]["Authorization"] == "Bearer sekrit"
    assert captured["verify"] == str(ca)
    c.close()


def test_from_kubeconfig_exec_plugin(tmp_path):
    """ExecCredential plugin auth: the plugin's stdout token is used."""
    plugin = tmp_path / "plugin.sh"
    plugin.write_text(
        "#!/bin/sh\n"
        "echo '{\"apiVersion\":\"client.authentication.k8s.io/v1\","
        "\"kind\":\"ExecCredential\",\"status\":{\"token\":\"exec-token\"}}'\n")
    plugin.chmod(0o755)
    cfg = {
        "current-context": "c",
        "contexts": [{"name": "c", "context": {"cluster": "cl", "user": "u"}}],
        "clusters": [{"name": "cl", "cluster": {
            "server": "https://h:6443", "insecure-skip-tls-verify": True}}],
        "users": [{"name": "u", "user": {"exec": {
            "apiVersion": "client.authentication.k8s.io/v1",
            "command": str(plugin)}}}],
    }
    captured = {}

    class CapturingClient:
        def __init__(self, **kw):
            captured.update(kw)

        def close(self):
            pass

    real_httpx_client = httpx.Client
    httpx.Client = lambda **kw: CapturingClient(**kw)
    try:
        c = RealKubeClient.from_kubeconfig(cfg)
    finally:
        httpx.Client = real_httpx_client
    assert captured["headers"]["Authorization"] == "Bearer exec-token"
    assert captured["verify"] is False
    c.close()
