#!/bin/bash
set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
timeout 300 python -m pytest tests/test_ops_gpu.py -x -q -m gpu 2>&1 | tail -3
timeout 700 python bench.py --steps 3 --warmup 1 > gpurun_out/bench_flagship.json 2> gpurun_out/bench_flagship.log
echo "graphs-on rc=$?"; cat gpurun_out/bench_flagship.json | head -c 300; echo; grep stage gpurun_out/bench_flagship.log
SENTIO_DISABLE_HIPGRAPH=1 timeout 700 python bench.py --steps 2 --warmup 1 > gpurun_out/bench_nograph.json 2> gpurun_out/bench_nograph.log
echo "graphs-off rc=$?"; cat gpurun_out/bench_nograph.json | head -c 300; echo; grep stage gpurun_out/bench_nograph.log
export TMPDIR=/tmp; cd /tmp
timeout 500 rocprofv3 --kernel-trace --stats --output-format csv -d /tmp/prof -o bench -- python "$GRAFT_REPO_ROOT/bench.py" --steps 2 --warmup 1 --docs-per-gpu 300000 --batch 16 --gen-tokens 64 > "$GRAFT_REPO_ROOT/gpurun_out/prof_run.log" 2>&1
cp /tmp/prof/bench_kernel_stats.csv "$GRAFT_REPO_ROOT/gpurun_out/" 2>/dev/null
SENTIO_DISABLE_HIPGRAPH=1 timeout 500 rocprofv3 --kernel-trace --stats --output-format csv -d /tmp/prof2 -o bench -- python "$GRAFT_REPO_ROOT/bench.py" --steps 2 --warmup 1 --docs-per-gpu 300000 --batch 16 --gen-tokens 64 > "$GRAFT_REPO_ROOT/gpurun_out/prof_nograph.log" 2>&1
cp /tmp/prof2/bench_kernel_stats.csv "$GRAFT_REPO_ROOT/gpurun_out/bench_kernel_stats_nograph.csv" 2>/dev/null
echo done
