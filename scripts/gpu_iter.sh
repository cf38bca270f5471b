#!/bin/bash
set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
timeout 420 python -m pytest tests -x -q -m gpu 2>&1 | tail -6
timeout 700 python bench.py --steps 3 --warmup 1 > gpurun_out/bench_flagship.json 2> gpurun_out/bench_flagship.log
echo "bench rc=$?"; cat gpurun_out/bench_flagship.json; tail -5 gpurun_out/bench_flagship.log
export TMPDIR=/tmp; cd /tmp
timeout 500 rocprofv3 --kernel-trace --stats --output-format csv -d /tmp/prof -o bench -- python "$GRAFT_REPO_ROOT/bench.py" --steps 2 --warmup 1 --docs-per-gpu 300000 --batch 16 --gen-tokens 64 > "$GRAFT_REPO_ROOT/gpurun_out/prof_run.log" 2>&1
echo "prof rc=$?"
cp /tmp/prof/bench_kernel_stats.csv "$GRAFT_REPO_ROOT/gpurun_out/" 2>/dev/null
grep -m1 '"metric"' "$GRAFT_REPO_ROOT/gpurun_out/prof_run.log" | head -c 400
