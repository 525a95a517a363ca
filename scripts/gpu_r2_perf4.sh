#!/bin/bash
set -x
mkdir -p gpurun_out
timeout 500 python bench.py --steps 200 --warmup 5 --kube-sim > gpurun_out/r2_kube_sim_soak.json 2>/dev/null
python -c "import json; d=json.load(open('gpurun_out/r2_kube_sim_soak.json')); c=d['config']; print('kube-sim:', d['value'], 'p50', c['p50_filter_bind_ms'], 'retries', c['bind_retries'])"
timeout 400 python bench.py --steps 50 --warmup 3 --nodes 2 --cards 64 --batch 128 > gpurun_out/r2_cpx64.json 2>/dev/null
python -c "import json; d=json.load(open('gpurun_out/r2_cpx64.json')); c=d['config']; print('CPX-64:', d['value'], 'p50', c['p50_filter_bind_ms'])"
timeout 400 python bench.py --steps 100 --warmup 5 > gpurun_out/r2_plain_ab.json 2>/dev/null
timeout 400 python bench.py --steps 100 --warmup 5 --tls > gpurun_out/r2_tls_ab.json 2>/dev/null
python - <<'PY'
import json
for name in ("plain", "tls"):
    d = json.load(open(f"gpurun_out/r2_{name}_ab.json"))
    c = d["config"]
    print(name, d["value"], "pods/s p50", c["p50_filter_bind_ms"], "native filters", c["native_stats"]["filter_native"])
PY
