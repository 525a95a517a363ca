#!/bin/bash
# Round-2 perf validation on MI355X box: gpu tests, default bench,
# 256-node A/B vs r1's 411 pods/s, policy soaks with de-herding,
# rocprof kernel-trace evidence.
set -x
mkdir -p gpurun_out
nproc
python -m pytest tests -m gpu -q 2>&1 | tail -3
timeout 500 python bench.py --steps 100 --warmup 5 > gpurun_out/r2_bench_default.json 2>/dev/null
tail -c 1200 gpurun_out/r2_bench_default.json
timeout 500 python bench.py --steps 50 --warmup 3 --nodes 256 --batch 128 > gpurun_out/r2_bench_256nodes.json 2>/dev/null
tail -c 1200 gpurun_out/r2_bench_256nodes.json
timeout 600 python bench.py --steps 200 --warmup 5 --policy binpack > gpurun_out/r2_soak_binpack.json 2>/dev/null
tail -c 1200 gpurun_out/r2_soak_binpack.json
timeout 600 python bench.py --steps 200 --warmup 5 --policy spread > gpurun_out/r2_soak_spread.json 2>/dev/null
tail -c 1200 gpurun_out/r2_soak_spread.json
cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT"
(cd /tmp && rocprofv3 --kernel-trace --stats -d /tmp/prof_r2 -o r2trace -- bash -c "cd $GRAFT_REPO_ROOT && timeout 240 python bench.py --steps 10 --warmup 2 > gpurun_out/r2_bench_traced.json 2>/dev/null") 2>&1 | tail -3
find /tmp/prof_r2 -name '*.csv' | head -5
cp /tmp/prof_r2/*/*kernel*csv gpurun_out/ 2>/dev/null || cp /tmp/prof_r2/*kernel*csv gpurun_out/ 2>/dev/null || find /tmp/prof_r2 -name '*.csv' -exec cp {} gpurun_out/ \;
ls gpurun_out/
