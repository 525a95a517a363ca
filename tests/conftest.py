"""Shared fixtures. Registers the `gpu` marker: tests needing a real MI355X
are marked @pytest.mark.gpu and run on GPU boxes only; everything else runs
on CPU."""
from __future__ import annotations

import sys
import uuid
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent
if str(REPO) not in sys.path:
    sys.path.insert(0, str(REPO))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: requires a real MI355X GPU (run via gpurun)")


def pytest_collection_modifyitems(config, items):
    """Skip gpu-marked tests automatically when no GPU is visible, unless the
    run explicitly selects them with -m gpu."""
    import torch

    if torch.cuda.is_available():
        return
    marker_expr = config.getoption("-m") or ""
    skip = pytest.mark.skip(reason="no GPU visible")
    for item in items:
        if "gpu" in item.keywords and "gpu" not in marker_expr:
            item.add_marker(skip)


GiB = 1024**3


def make_node(name: str, cards: int = 8, mem_per_card: int = 288 * GiB,
              annotations: dict | None = None) -> dict:
    node = {
        "metadata": {"name": name},
        "status": {"allocatable": {
            "elasticgpu.io/gpu-core": str(100 * cards),
            "elasticgpu.io/gpu-memory": str(mem_per_card * cards),
        }},
    }
    if annotations:
        node["metadata"]["annotations"] = annotations
    return node


def make_pod(name: str, core: int = 0, memory: int = 0, containers: int = 1,
             namespace: str = "default", uid: str | None = None,
             per_container: list[dict] | None = None) -> dict:
    """A GPU pod spec. Either uniform (core/memory per container) or
    per_container=[{"core":..,"memory":..}, ...]."""
    specs = per_container or [{"core": core, "memory": memory}] * containers
    conts = []
    for i, s in enumerate(specs):
        req = {}
        if s.get("core"):
            req["elasticgpu.io/gpu-core"] = str(s["core"])
        if s.get("memory"):
            req["elasticgpu.io/gpu-memory"] = str(s["memory"])
        if s.get("pgpu"):
            req["elasticgpu.io/pgpu"] = str(s["pgpu"])
        conts.append({"name": f"c{i}",
                      "resources": {"requests": req, "limits": dict(req)}})
    pod = {
        "metadata": {"name": name, "namespace": namespace,
                     "uid": uid or str(uuid.uuid4())},
        "spec": {"containers": conts},
        "status": {"phase": "Pending"},
    }
    return pod


@pytest.fixture
def fake_client():
    from elastic_gpu_scheduler_amd.k8s.client import FakeKubeClient

    return FakeKubeClient()


@pytest.fixture
def cluster(fake_client):
    """A fake 2-node 8x MI355X cluster with a binpack registry + ASGI app."""
    from elastic_gpu_scheduler_amd.scheduler.service import SchedulerRegistry
    from elastic_gpu_scheduler_amd.server.app import make_app

    fake_client.add_node(make_node("node-a"))
    fake_client.add_node(make_node("node-b"))
    registry = SchedulerRegistry(fake_client)
    app = make_app(registry)
    return fake_client, registry, app


class ExtenderClient:
    """Synchronous helper around the ASGI app for protocol tests."""

    def __init__(self, app):
        import httpx

        self._transport = httpx.ASGITransport(app=app)
        self._base = "http://egs"

    def request(self, method: str, path: str, json_body=None, content=None):
        import asyncio

        import httpx

        async def go():
            async with httpx.AsyncClient(transport=self._transport,
                                         base_url=self._base) as c:
                return await c.request(method, path, json=json_body,
                                       content=content)

        return asyncio.run(go())

    def filter(self, pod, nodes):
        return self.request("POST", "/scheduler/filter",
                            {"pod": pod, "nodenames": nodes})

    def priorities(self, pod, nodes):
        return self.request("POST", "/scheduler/priorities",
                            {"pod": pod, "nodenames": nodes})

    def bind(self, pod, node):
        return self.request("POST", "/scheduler/bind", {
            "podName": pod["metadata"]["name"],
            "podNamespace": pod["metadata"]["namespace"],
            "podUID": pod["metadata"]["uid"],
            "node": node,
        })


@pytest.fixture
def extender(cluster):
    _, _, app = cluster
    return ExtenderClient(app)
