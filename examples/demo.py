#!/usr/bin/env python3
"""Five-minute tour: an in-process MI355X cluster, the native extender, and
every scheduling mode — run `python examples/demo.py` (no cluster, no GPU
needed; uses the fake apiserver).
"""
from __future__ import annotations

import json
import sys
import uuid
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from elastic_gpu_scheduler_amd.k8s.client import FakeKubeClient  # noqa: E402
from elastic_gpu_scheduler_amd.scheduler.service import SchedulerRegistry  # noqa: E402
from elastic_gpu_scheduler_amd.server.app import make_app  # noqa: E402
from elastic_gpu_scheduler_amd.server.native import NativeFrontend  # noqa: E402

GiB = 1024**3


def pod(name, resources, containers=1, annotations=None):
    conts = [{"name": f"c{i}",
              "resources": {"requests": {k: str(v) for k, v in resources.items()}}}
             for i in range(containers)]
    meta = {"name": name, "namespace": "default", "uid": str(uuid.uuid4())}
    if annotations:
        meta["annotations"] = annotations
    return {"metadata": meta, "spec": {"containers": conts},
            "status": {"phase": "Pending"}}


def main():
    client = FakeKubeClient()
    # two 8x MI355X nodes; node-b is a partitioned hive (two xGMI islands)
    hops_b = [[0 if i == j else (1 if (i < 4) == (j < 4) else 3)
               for j in range(8)] for i in range(8)]
    for name, ann in (("node-a", None),
                      ("node-b", {"elasticgpu.io/xgmi-topology":
                                  json.dumps({"hops": hops_b})})):
        node = {"metadata": {"name": name},
                "status": {"allocatable": {
                    "elasticgpu.io/gpu-core": "800",
                    "elasticgpu.io/gpu-memory": str(8 * 288 * GiB)}}}
        if ann:
            node["metadata"]["annotations"] = ann
        client.add_node(node)

    registry = SchedulerRegistry(client, policy="binpack")
    fe = NativeFrontend(make_app(registry), host="127.0.0.1", port=0)
    fe.start()
    print(f"native extender serving on 127.0.0.1:{fe.port}\n")

    import httpx

    c = httpx.Client(base_url=f"http://127.0.0.1:{fe.port}", timeout=10)
    nodes = ["node-a", "node-b"]

    def schedule(p, label):
        created = client.create_pod(p)
        ok = c.post("/scheduler/filter",
                    json={"pod": created, "nodenames": nodes}).json()
        prio = c.post("/scheduler/priorities",
                      json={"pod": created,
                            "nodenames": ok["nodenames"]}).json()
        best = max(prio, key=lambda e: e["score"])["host"]
        r = c.post("/scheduler/bind", json={
            "podName": created["metadata"]["name"], "podNamespace": "default",
            "podUID": created["metadata"]["uid"], "node": best})
        assert r.status_code == 200, r.text
        bound = client.get_pod("default", created["metadata"]["name"])
        cards = [v for k, v in bound["metadata"]["annotations"].items()
                 if k.startswith("elasticgpu.io/container-")]
        print(f"{label:46s} -> {best} cards {cards}")

    schedule(pod("frac-quarter", {"elasticgpu.io/gpu-core": 25,
                                  "elasticgpu.io/gpu-memory": "64Gi"}),
             "fractional: 25% core + 64Gi")
    schedule(pod("mem-only", {"elasticgpu.io/gpu-memory": "100Gi"}),
             "memory-only share: 100Gi")
    schedule(pod("whole-one", {"elasticgpu.io/gpu-core": 100}),
             "whole card: gpu-core=100")
    schedule(pod("multi-4", {"elasticgpu.io/gpu-core": 400}),
             "4 cards: gpu-core=400 (xGMI-adjacent set)")
    schedule(pod("pgpu-2", {"elasticgpu.io/pgpu": 2}),
             "pgpu: 2 exclusive cards")
    schedule(pod("spread-3", {"elasticgpu.io/gpu-core": 20}, containers=3,
                 annotations={"elasticgpu.io/spread-containers": "true"}),
             "3 containers spread to distinct cards")

    print("\n/scheduler/status excerpt:")
    st = c.get("/scheduler/status").json()
    for n, info in st["gpushare"]["nodes"].items():
        used = [f"{100 - g['core_available']}%" for g in info["gpus"]]
        print(f"  {n}: core used per card {used}")
    print("\nnative fast-path stats:", fe.stats())
    c.close()
    fe.stop()


if __name__ == "__main__":
    main()
