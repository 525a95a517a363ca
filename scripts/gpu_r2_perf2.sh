#!/bin/bash
# Round-2 GPU batch 2: rocprof kernel evidence, 256-node concurrency
# sweep, marathon soak, real-wire e2e on the 256-core box.
set -x
mkdir -p gpurun_out
cd /tmp && export TMPDIR=/tmp
export PYTHONPATH=/root/repo
rocprofv3 --kernel-trace --stats -d /tmp/prof_r2 -o r2trace -- python /root/repo/bench.py --steps 5 --warmup 1 --verify-sample 8 > /tmp/r2_traced_stdout.json 2>/tmp/r2_rocprof.err
tail -c 300 /tmp/r2_traced_stdout.json
tail -5 /tmp/r2_rocprof.err
find /tmp/prof_r2 -type f | head -10
for f in $(find /tmp/prof_r2 -name '*kernel*' -o -name '*stats*' | head -6); do cp "$f" /root/repo/gpurun_out/r2_$(basename $f); done
cd /root/repo
for c in 4 8 16 32; do
  timeout 400 python bench.py --steps 30 --warmup 3 --nodes 256 --batch 128 --concurrency $c 2>/dev/null | tail -1 >> gpurun_out/r2_256n_conc_sweep.jsonl
done
cat gpurun_out/r2_256n_conc_sweep.jsonl | python -c "
import json,sys
for l in sys.stdin:
    d=json.loads(l); c=d['config']
    print('conc', c['concurrency'], d['value'], 'pods/s p50', c['p50_filter_bind_ms'])"
timeout 900 python bench.py --steps 2000 --warmup 10 > gpurun_out/r2_marathon_128k.json 2>/dev/null
tail -c 700 gpurun_out/r2_marathon_128k.json
timeout 500 python benchmarks/e2e_real_wire.py --steps 5 --warmup 2 --batch 200 > gpurun_out/r2_real_wire_gpu_box.json 2>/dev/null
cat gpurun_out/r2_real_wire_gpu_box.json
