#!/bin/bash
# Round-2 consolidated GPU validation: full gpu suite + smoke + all
# headline benches on the final build + sweep + fanout + real-wire 8x8.
set -x
mkdir -p gpurun_out
python -m pytest tests -m gpu -q 2>&1 | tail -3
python - <<'PY'
import __graft_entry__ as g
g.smoke()
print("SMOKE_OK")
PY
run() { name=$1; shift; timeout 700 python bench.py "$@" > gpurun_out/r2f_$name.json 2>/dev/null; \
  python -c "import json; d=json.load(open('gpurun_out/r2f_$name.json')); c=d['config']; print('$name:', d['value'], 'pods/s p50', c['p50_filter_bind_ms'], 'p99', c['p99_filter_bind_ms'], 'retries', c['bind_retries'])"; }
run default --steps 200 --warmup 5
run kube_sim --steps 200 --warmup 5 --kube-sim
run 256nodes --steps 50 --warmup 3 --nodes 256 --batch 128
run cpx64 --steps 50 --warmup 3 --nodes 2 --cards 64 --batch 128
run tls --steps 100 --warmup 5 --tls
timeout 900 python bench.py --steps 5000 --warmup 10 > gpurun_out/r2f_marathon_320k.json 2>/dev/null
python -c "import json; d=json.load(open('gpurun_out/r2f_marathon_320k.json')); c=d['config']; print('marathon 320k:', d['value'], 'pods/s p50', c['p50_filter_bind_ms'], 'retries', c['bind_retries'])"
timeout 500 python benchmarks/sweep.py --json gpurun_out/r2f_sweep.json 2>&1 | tail -2
timeout 500 python benchmarks/fanout.py --json gpurun_out/r2f_fanout.json 2>/dev/null | tail -3
timeout 500 python benchmarks/e2e_real_wire.py --steps 5 --warmup 2 --batch 64 --nodes 8 --cards 8 > gpurun_out/r2f_real_wire_8x8.json 2>/dev/null
cat gpurun_out/r2f_real_wire_8x8.json
