#!/bin/bash
set -x
mkdir -p gpurun_out
cd /tmp && export TMPDIR=/tmp
export PYTHONPATH=/root/repo
rocprofv3 --kernel-trace --stats --output-format csv -d /tmp/prof_r2 -o r2trace -- python /root/repo/bench.py --steps 5 --warmup 1 --verify-sample 8 > /tmp/r2_traced_stdout.json 2>/tmp/r2_rocprof.err
find /tmp/prof_r2 -type f | head
for f in $(find /tmp/prof_r2 -name '*.csv' | head -8); do cp "$f" /root/repo/gpurun_out/r2_$(basename $f); done
tail -5 /tmp/r2_rocprof.err
cd /root/repo
timeout 400 python bench.py --steps 50 --warmup 3 --nodes 256 --batch 128 > gpurun_out/r2_256n_inline_cache.json 2>/dev/null
python -c "import json; d=json.load(open('gpurun_out/r2_256n_inline_cache.json')); c=d['config']; print('256n inline-cache:', d['value'], 'p50', c['p50_filter_bind_ms'], 'p99', c['p99_filter_bind_ms'])"
timeout 600 python bench.py --steps 200 --warmup 5 --policy binpack --pod-pad-bytes 8192 > gpurun_out/r2_soak_binpack_8kb.json 2>/dev/null
python -c "import json; d=json.load(open('gpurun_out/r2_soak_binpack_8kb.json')); c=d['config']; print('binpack+8KB pods:', d['value'], 'p50', c['p50_filter_bind_ms'], 'retries', c['bind_retries'])"
