#!/bin/bash
set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
timeout 420 python -m pytest tests -x -q -m gpu 2>&1 | tail -2
timeout 600 python bench.py --steps 4 --warmup 1 > gpurun_out/bench_a.json 2> gpurun_out/bench_a.log
grep -o '"value": [0-9.]*' gpurun_out/bench_a.json | head -1; grep stage gpurun_out/bench_a.log
timeout 600 python bench.py --steps 4 --warmup 1 > gpurun_out/bench_b.json 2> gpurun_out/bench_b.log
grep -o '"value": [0-9.]*' gpurun_out/bench_b.json | head -1; grep stage gpurun_out/bench_b.log
cp gpurun_out/bench_a.json gpurun_out/bench_flagship.json
export TMPDIR=/tmp; cd /tmp
timeout 500 rocprofv3 --kernel-trace --stats --output-format csv -d /tmp/prof -o bench -- python "$GRAFT_REPO_ROOT/bench.py" --steps 2 --warmup 1 --docs-per-gpu 300000 --batch 16 --gen-tokens 64 > "$GRAFT_REPO_ROOT/gpurun_out/prof_run.log" 2>&1
cp /tmp/prof/bench_kernel_stats.csv "$GRAFT_REPO_ROOT/gpurun_out/" 2>/dev/null
echo done
