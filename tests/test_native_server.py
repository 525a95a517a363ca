"""Native C++ HTTP server: wire parity with the Python ASGI app, fallback
routing, keep-alive, and the JSON codec."""
from __future__ import annotations

import json
import socket

import httpx
import pytest

from elastic_gpu_scheduler_amd._native import core
from elastic_gpu_scheduler_amd.scheduler.service import SchedulerRegistry
from elastic_gpu_scheduler_amd.server.app import make_app
from elastic_gpu_scheduler_amd.server.native import NativeFrontend
from tests.conftest import make_node, make_pod

GiB = 1024**3


@pytest.fixture
def native(fake_client):
    fake_client.add_node(make_node("node-a"))
    fake_client.add_node(make_node("node-b"))
    registry = SchedulerRegistry(fake_client)
    app = make_app(registry)
    fe = NativeFrontend(app, host="127.0.0.1", port=0)
    fe.start()
    yield fake_client, registry, fe
    fe.stop()


def _client(fe):
    return httpx.Client(base_url=f"http://127.0.0.1:{fe.port}", timeout=10.0)


def test_json_codec_roundtrip():
    cases = [
        '{"a": 1, "b": [true, false, null], "c": {"d": "x\\ny"}}',
        '{"nested": {"deep": {"deeper": [1.5, -2, 1e3]}}}',
        '{"uni": "\\u00e9\\u4e2d\\ud83d\\ude00"}',
        '[]', '{}', '"plain"', '42', '-3.25', 'true', 'null',
    ]
    for case in cases:
        out = core.json_roundtrip(case)
        assert json.loads(out) == json.loads(case), case
    with pytest.raises(RuntimeError):
        core.json_roundtrip("{broken")


def test_native_filter_matches_python_app(native):
    client, registry, fe = native
    pod = client.create_pod(make_pod("p", core=30, memory=64 * GiB))
    # warm the node cache through the python path first
    registry.default.assume(["node-a", "node-b"], pod)
    with _client(fe) as c:
        r = c.post("/scheduler/filter",
                   json={"pod": pod, "nodenames": ["node-a", "node-b"]})
        assert r.status_code == 200
        assert sorted(r.json()["nodenames"]) == ["node-a", "node-b"]
        stats = fe.stats()
        assert stats["filter_native"] == 1

        r = c.post("/scheduler/priorities",
                   json={"pod": pod, "nodenames": ["node-a", "node-b"]})
        scores = r.json()
        assert {e["host"] for e in scores} == {"node-a", "node-b"}
        assert all(isinstance(e["score"], int) for e in scores)
        assert fe.stats()["priorities_native"] == 1


def test_native_cold_node_falls_back_then_warms(native):
    client, registry, fe = native
    pod = client.create_pod(make_pod("p", core=30, memory=64 * GiB))
    with _client(fe) as c:
        r = c.post("/scheduler/filter",
                   json={"pod": pod, "nodenames": ["node-a"]})
        assert r.status_code == 200 and r.json()["nodenames"] == ["node-a"]
        assert fe.stats()["fallback"] >= 1  # cold cache -> python
        r = c.post("/scheduler/filter",
                   json={"pod": pod, "nodenames": ["node-a"]})
        assert fe.stats()["filter_native"] == 1  # now native


def test_native_full_schedule_cycle(native):
    client, registry, fe = native
    pod = client.create_pod(make_pod("p", core=25, memory=48 * GiB))
    with _client(fe) as c:
        r = c.post("/scheduler/filter",
                   json={"pod": pod, "nodenames": ["node-a", "node-b"]})
        ok = r.json()["nodenames"]
        r = c.post("/scheduler/priorities", json={"pod": pod, "nodenames": ok})
        best = max(r.json(), key=lambda e: e["score"])["host"]
        r = c.post("/scheduler/bind", json={
            "podName": "p", "podNamespace": "default",
            "podUID": pod["metadata"]["uid"], "node": best})
        assert r.status_code == 200
    bound = client.get_pod("default", "p")
    assert bound["spec"]["nodeName"] == best
    assert bound["metadata"]["annotations"]["elasticgpu.io/assumed"] == "true"


def test_native_bad_json_gets_python_400(native):
    _, _, fe = native
    with _client(fe) as c:
        r = c.post("/scheduler/filter", content=b"{nope",
                   headers={"content-type": "application/json"})
        assert r.status_code == 400


def test_native_get_routes_fall_back(native):
    client, registry, fe = native
    with _client(fe) as c:
        assert c.get("/healthz").json() == {"ok": True}
        assert c.get("/version").json()["target"].startswith("MI355X")
        assert b"egs_requests_total" in c.get("/metrics").content
        assert c.get("/nope").status_code == 404


def test_native_keep_alive_many_requests(native):
    client, registry, fe = native
    pod = client.create_pod(make_pod("p", core=10, memory=GiB))
    registry.default.assume(["node-a"], pod)
    body = json.dumps({"pod": pod, "nodenames": ["node-a"]}).encode()
    payload = (b"POST /scheduler/filter HTTP/1.1\r\nhost: t\r\n"
               b"content-length: " + str(len(body)).encode() + b"\r\n\r\n" + body)
    s = socket.create_connection(("127.0.0.1", fe.port))
    try:
        for _ in range(50):
            s.sendall(payload)
            data = b""
            while b"\r\n\r\n" not in data:
                data += s.recv(65536)
            head, _, rest = data.partition(b"\r\n\r\n")
            clen = int([l for l in head.split(b"\r\n")
                        if l.lower().startswith(b"content-length")][0]
                       .split(b":")[1])
            while len(rest) < clen:
                rest += s.recv(65536)
            out = json.loads(rest[:clen])
            assert out["nodenames"] == ["node-a"]
    finally:
        s.close()
    assert fe.stats()["filter_native"] >= 50


def test_native_filter_to_bind_latency_recorded(native):
    client, registry, fe = native
    pod = client.create_pod(make_pod("p", core=25, memory=GiB))
    registry.default._ensure_node("node-a")
    with _client(fe) as c:
        c.post("/scheduler/filter", json={"pod": pod, "nodenames": ["node-a"]})
        assert fe.stats()["filter_native"] == 1
        r = c.post("/scheduler/bind", json={
            "podName": "p", "podNamespace": "default",
            "podUID": pod["metadata"]["uid"], "node": "node-a"})
        assert r.status_code == 200
    # the native tracker entry was consumed by the bind handler
    assert fe.server.pop_filter_seconds(pod["metadata"]["uid"]) < 0


def test_native_http_edge_cases(native):
    """Connection: close honored; bad request line closes; metrics include
    native counters."""
    client, registry, fe = native
    import socket as _socket

    # Connection: close
    s = _socket.create_connection(("127.0.0.1", fe.port))
    s.sendall(b"GET /healthz HTTP/1.1\r\nhost: t\r\nconnection: close\r\n\r\n")
    data = b""
    while True:
        chunk = s.recv(65536)
        if not chunk:
            break
        data += chunk
    assert b'{"ok": true}' in data
    assert b"connection: close" in data.lower()
    s.close()

    # malformed request line -> server closes without crashing
    s = _socket.create_connection(("127.0.0.1", fe.port))
    s.sendall(b"BOGUS\r\n\r\n")
    s.settimeout(5)
    assert s.recv(1024) == b""  # closed
    s.close()

    # oversized content-length refused (connection closed; cap is 8 MiB)
    s = _socket.create_connection(("127.0.0.1", fe.port))
    s.sendall(b"POST /scheduler/filter HTTP/1.1\r\nhost: t\r\n"
              b"content-length: 99999999\r\n\r\n")
    s.settimeout(5)
    assert s.recv(1024) == b""
    s.close()

    # server still serves after the bad clients
    with _client(fe) as c:
        assert c.get("/healthz").status_code == 200
        m = c.get("/metrics").content
        assert b"egs_native_requests_total" in m


def test_native_honors_spread_containers(native):
    """The C++ fast path must apply the spread-containers constraint too."""
    client, registry, fe = native
    registry.default._ensure_node("node-a")
    pod = client.create_pod({
        "metadata": {"name": "sp", "namespace": "default", "uid": "sp-uid",
                     "annotations": {"elasticgpu.io/spread-containers": "true"}},
        "spec": {"containers": [
            {"name": f"c{i}", "resources": {"requests": {
                "elasticgpu.io/gpu-core": "20"}}} for i in range(3)]},
        "status": {"phase": "Pending"}})
    with _client(fe) as c:
        r = c.post("/scheduler/filter",
                   json={"pod": pod, "nodenames": ["node-a"]})
        assert r.json()["nodenames"] == ["node-a"]
        assert fe.stats()["filter_native"] == 1
        r = c.post("/scheduler/bind", json={
            "podName": "sp", "podNamespace": "default", "podUID": "sp-uid",
            "node": "node-a"})
        assert r.status_code == 200
    from elastic_gpu_scheduler_amd.k8s import objects as obj
    alloc = obj.parse_allocation(client.get_pod("default", "sp"))
    assert len({a[0] for a in alloc}) == 3


def test_native_connection_churn_and_partial_requests(native):
    """Abrupt disconnects, partial requests, and many short-lived
    connections must not wedge or leak the server."""
    import socket as _socket
    import threading

    client, registry, fe = native
    pod = client.create_pod(make_pod("p", core=10, memory=GiB))
    registry.default.assume(["node-a"], pod)
    body = json.dumps({"pod": pod, "nodenames": ["node-a"]}).encode()
    payload = (b"POST /scheduler/filter HTTP/1.1\r\nhost: t\r\n"
               b"content-length: " + str(len(body)).encode() + b"\r\n\r\n" + body)

    def churn(kind):
        for _ in range(30):
            s = _socket.create_connection(("127.0.0.1", fe.port))
            try:
                if kind == "abrupt":
                    s.sendall(payload[: len(payload) // 2])  # die mid-request
                elif kind == "headers-only":
                    s.sendall(b"POST /scheduler/filter HTTP/1.1\r\n"
                              b"content-length: 100\r\n\r\n")  # body never comes
                else:
                    s.sendall(payload)
                    data = b""
                    while b"\r\n\r\n" not in data:
                        data += s.recv(65536)
            finally:
                s.close()

    threads = [threading.Thread(target=churn, args=(k,))
               for k in ("abrupt", "headers-only", "full", "full")]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=60)
    assert not any(t.is_alive() for t in threads)
    # server still healthy afterwards
    with _client(fe) as c:
        assert c.get("/healthz").status_code == 200
        r = c.post("/scheduler/filter",
                   json={"pod": pod, "nodenames": ["node-a"]})
        assert r.status_code == 200


def test_nesting_bomb_rejected_not_crashed(native):
    """A deeply-nested JSON body must produce a 4xx, crash neither the C++
    parser (stack-depth cap) nor the Python fallback (RecursionError ->
    400)."""
    _, _, fe = native
    with _client(fe) as c:
        r = c.post("/scheduler/filter", content=b"[" * 500000,
                   headers={"content-type": "application/json"})
        assert r.status_code == 400
        assert c.get("/healthz").status_code == 200
    assert core.json_roundtrip("[" * 200 + "1" + "]" * 200)
    with pytest.raises(RuntimeError):
        core.json_roundtrip("[" * 1000 + "1" + "]" * 1000)


def test_native_latency_histograms_visible(native):
    """the GIL-free fast path exports its own per-verb
    latency histograms (C++-computed), surfaced on /metrics."""
    client, registry, fe = native
    pod = client.create_pod(make_pod("h", core=10, memory=GiB))
    with _client(fe) as c:
        for _ in range(20):
            c.post("/scheduler/filter",
                   json={"pod": pod, "nodenames": ["node-a"]})
            c.post("/scheduler/priorities",
                   json={"pod": pod, "nodenames": ["node-a"]})
    hists = fe.server.latency_histograms()
    for verb in ("filter", "priorities"):
        h = hists[verb]
        assert h["count"] >= 20
        assert h["sum_us"] > 0
        assert sum(n for _, n in h["buckets"]) == h["count"]
    # rendered on /metrics as a cumulative Prometheus histogram
    st, _, body = fe.app.handle("GET", "/metrics", b"")
    text = body.decode()
    assert st == 200
    assert 'egs_native_verb_latency_seconds_bucket{verb="filter"' in text
    assert 'egs_native_verb_latency_seconds_count{verb="priorities"}' in text
    assert 'le="+Inf"' in text


def test_debug_latency_and_heap_endpoints(native):
    client, registry, fe = native
    pod = client.create_pod(make_pod("d", core=10, memory=GiB))
    with _client(fe) as c:
        for _ in range(5):
            c.post("/scheduler/filter",
                   json={"pod": pod, "nodenames": ["node-a"]})
        r = c.get("/debug/latency")
        assert r.status_code == 200
        d = r.json()
        assert d["native"]["filter"]["count"] >= 5
        assert d["native"]["filter"]["p50_us"] is not None
        assert "filter" in d["python_verbs"] or d["python_verbs"] == {}
        # heap: first call arms tracemalloc, second returns a snapshot
        r1 = c.get("/debug/heap")
        assert r1.status_code == 200
        r2 = c.get("/debug/heap")
        assert r2.status_code == 200
        snap = r2.json()
        assert snap["rss_kib"] > 0
        assert isinstance(snap["top"], list) and snap["top"]


def test_native_server_ipv6(fake_client):
    """Dual-stack: an IPv6 host literal binds AF_INET6 (IPv6-first
    clusters); IPv4 behavior is unchanged (every other test)."""
    fake_client.add_node(make_node("node-a"))
    registry = SchedulerRegistry(fake_client)
    registry.default._ensure_node("node-a")
    app = make_app(registry)
    fe = NativeFrontend(app, host="::1", port=0)
    fe.start()
    try:
        with httpx.Client(base_url=f"http://[::1]:{fe.port}",
                          timeout=10.0) as c:
            assert c.get("/healthz").status_code == 200
            pod = fake_client.create_pod(make_pod("p6", core=10))
            r = c.post("/scheduler/filter",
                       json={"pod": pod, "nodenames": ["node-a"]})
            assert r.status_code == 200
            assert r.json()["nodenames"] == ["node-a"]
    finally:
        fe.stop()
