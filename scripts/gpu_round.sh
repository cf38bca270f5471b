#!/bin/bash
set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
echo "=== GPU tests ==="
timeout 420 python -m pytest tests -m gpu -x -q 2>&1 | tail -15
echo "=== bench flagship 1GPU ==="
timeout 600 python bench.py --steps 3 --warmup 1 > gpurun_out/bench_flagship.json 2> gpurun_out/bench_flagship.log
tail -2 gpurun_out/bench_flagship.json gpurun_out/bench_flagship.log
echo "=== rocprof stats (short bench) ==="
export TMPDIR=/tmp
cd /tmp
timeout 600 rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT/gpurun_out/prof" -o bench -- python "$GRAFT_REPO_ROOT/bench.py" --steps 2 --warmup 1 --docs-per-gpu 300000 --batch 16 --gen-tokens 64 > "$GRAFT_REPO_ROOT/gpurun_out/prof_run.log" 2>&1
tail -3 "$GRAFT_REPO_ROOT/gpurun_out/prof_run.log"
ls -la "$GRAFT_REPO_ROOT/gpurun_out/prof" 2>/dev/null | head
