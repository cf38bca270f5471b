#!/bin/bash
set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
timeout 420 python -m pytest tests -q -m gpu 2>&1 | tail -2
timeout 120 python -c "import __graft_entry__ as g; g.smoke()" 2>&1 | tail -1
timeout 600 python bench.py --steps 5 --warmup 2 > gpurun_out/bench_flagship.json 2> gpurun_out/bench_flagship.log
cat gpurun_out/bench_flagship.json | head -c 400; echo; grep stage gpurun_out/bench_flagship.log
export TMPDIR=/tmp; cd /tmp
timeout 400 rocprofv3 --kernel-trace --stats --output-format csv -d /tmp/prof -o bench -- python "$GRAFT_REPO_ROOT/bench.py" --steps 2 --warmup 1 --docs-per-gpu 300000 --batch 16 --gen-tokens 64 > "$GRAFT_REPO_ROOT/gpurun_out/prof_run.log" 2>&1
cp /tmp/prof/bench_kernel_stats.csv "$GRAFT_REPO_ROOT/gpurun_out/" 2>/dev/null
echo done
