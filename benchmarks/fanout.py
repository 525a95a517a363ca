#!/usr/bin/env python3
"""Filter fan-out scaling: one pod's feasibility checked across N nodes.

The reference fans the per-node check over a fixed 4-goroutine pool
(scheduler.go:135) regardless of cluster size; our ClusterState runs <=64
nodes inline and larger fan-outs on a pool sized to the host. This measures
a single assume() fan-out latency vs node count, native core only.

Run: python benchmarks/fanout.py [--json out.json]
"""
from __future__ import annotations

import argparse
import json
import statistics
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from elastic_gpu_scheduler_amd._native import core  # noqa: E402

GiB = 1024**3


def measure(n_nodes: int, iters: int = 200, threads: int = 0):
    cs = core.ClusterState("binpack", 0, threads)
    for i in range(n_nodes):
        cs.add_node(f"n{i}", [core.Device(100, 100, 288 * GiB, 288 * GiB)
                              for _ in range(8)], [])
    names = [f"n{i}" for i in range(n_nodes)]
    req = [core.GPUUnit(0, 25, 48 * GiB)]
    lats = []
    for it in range(iters):
        uid = f"pod-{it}"
        t0 = time.perf_counter()
        verdicts = cs.assume(names, uid, req)
        lats.append(time.perf_counter() - t0)
        assert all(v == 0 for v in verdicts)
    lats.sort()
    return {
        "nodes": n_nodes,
        "p50_us": round(statistics.median(lats) * 1e6, 1),
        "p99_us": round(lats[int(len(lats) * 0.99) - 1] * 1e6, 1),
        "per_node_ns": round(statistics.median(lats) / n_nodes * 1e9, 1),
    }


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--json", default="")
    args = p.parse_args()
    results = [measure(n) for n in (1, 8, 24, 64, 256, 1024, 4096)]
    for r in results:
        print(json.dumps(r), flush=True)
    if args.json:
        Path(args.json).write_text(json.dumps(results, indent=2))


if __name__ == "__main__":
    main()
