#!/bin/bash
set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
timeout 300 python -m pytest tests/test_ops_gpu.py -x -q -m gpu 2>&1 | tail -4
timeout 200 python scripts/gemm_probe.py > gpurun_out/gemm_probe.log 2>&1
cat gpurun_out/gemm_probe.log
timeout 700 python bench.py --steps 3 --warmup 1 > gpurun_out/bench_flagship.json 2> gpurun_out/bench_flagship.log
echo "bench rc=$?"; cat gpurun_out/bench_flagship.json | head -c 600; echo; tail -3 gpurun_out/bench_flagship.log
export TMPDIR=/tmp; cd /tmp
timeout 500 rocprofv3 --kernel-trace --stats --output-format csv -d /tmp/prof -o bench -- python "$GRAFT_REPO_ROOT/bench.py" --steps 2 --warmup 1 --docs-per-gpu 300000 --batch 16 --gen-tokens 64 > "$GRAFT_REPO_ROOT/gpurun_out/prof_run.log" 2>&1
cp /tmp/prof/bench_kernel_stats.csv "$GRAFT_REPO_ROOT/gpurun_out/" 2>/dev/null
echo prof_done
