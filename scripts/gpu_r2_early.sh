#!/bin/bash
# Round-2 early GPU validation: gpu tests, smoke, short bench, topology probe.
set -x
mkdir -p gpurun_out
python -m pytest tests -m gpu -q 2>&1 | tail -5
python - <<'PY'
import __graft_entry__ as g
g.smoke()
print("SMOKE_OK")
PY
timeout 400 python bench.py --steps 10 --warmup 2 > gpurun_out/bench_r2_early.json 2> gpurun_out/bench_r2_early.err
tail -c 2000 gpurun_out/bench_r2_early.json
python - <<'PY'
import json
from elastic_gpu_scheduler_amd.agent.agent import NodeAgent
a = NodeAgent("gpubox")
t = a.measured_topology()
h = a.health_check()
with open("gpurun_out/measured_topology_r2.json", "w") as f:
    json.dump({"topology": t, "health": h}, f, indent=1)
print("TOPO_OK", t["hops"], [round(x["hbm_gbps"]) for x in h])
PY
