#!/bin/bash
set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
echo "=== bench flagship 1GPU ==="
timeout 700 python bench.py --steps 3 --warmup 1 > gpurun_out/bench_flagship.json 2> gpurun_out/bench_flagship.log
echo "rc=$?"
cat gpurun_out/bench_flagship.json
echo "---- log ----"
tail -n 30 gpurun_out/bench_flagship.log
echo "=== rocprof stats (csv only) ==="
export TMPDIR=/tmp
cd /tmp
timeout 500 rocprofv3 --kernel-trace --stats --output-format csv -d /tmp/prof -o bench -- python "$GRAFT_REPO_ROOT/bench.py" --steps 2 --warmup 1 --docs-per-gpu 300000 --batch 16 --gen-tokens 64 > "$GRAFT_REPO_ROOT/gpurun_out/prof_run.log" 2>&1
echo "prof rc=$?"
tail -n 5 "$GRAFT_REPO_ROOT/gpurun_out/prof_run.log"
cp /tmp/prof/*kernel_stats.csv "$GRAFT_REPO_ROOT/gpurun_out/" 2>/dev/null
ls -la /tmp/prof
head -c 2000 "$GRAFT_REPO_ROOT/gpurun_out/"*kernel_stats.csv 2>/dev/null
