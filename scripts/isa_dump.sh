#!/usr/bin/env bash
# Dump gfx950 ISA for a kernel translation unit and summarize the things
# that decided every perf fight so far (profiles/README "ISA story"):
#   - scratch spills (scratch_size > 0 kills occupancy-sensitive kernels)
#   - VGPR/SGPR counts and occupancy
#   - s_waitcnt vmcnt(0) immediately after lone global_loads (the
#     zero-memory-level-parallelism signature of a dynamic trip count)
# Usage: scripts/isa_dump.sh sentio_amd/ops/csrc/attention.hip [out.s]
set -euo pipefail
TU=${1:?usage: isa_dump.sh <file.hip> [out.s]}
OUT=${2:-/tmp/$(basename "$TU" .hip).s}
ROOT=$(cd "$(dirname "$0")/.." && pwd)

TORCH_INC=$(python3 -c "import torch, os; p=os.path.dirname(torch.__file__); print(f'-I{p}/include -I{p}/include/torch/csrc/api/include')")
hipcc --offload-arch=gfx950 -O3 -std=c++17 -DNDEBUG -S $TORCH_INC \
      -I"$ROOT" -o "$OUT" "$TU"
echo "ISA: $OUT"
echo "--- resource usage per kernel ---"
grep -E "\.amdhsa_kernel|\.amdhsa_next_free_vgpr|\.amdhsa_next_free_sgpr|\.amdhsa_private_segment_fixed_size|\.amdhsa_accum_offset" "$OUT" \
  | sed 's/^\s*//'
echo "--- spill check (non-zero private_segment = scratch spills) ---"
grep -c "scratch_" "$OUT" | xargs echo "scratch refs:"
echo "--- serialized-load signature (load followed by vmcnt(0)) ---"
awk '/global_load|buffer_load/{l=NR} /s_waitcnt.*vmcnt\(0\)/{if (NR==l+1) n++} END{print n+0, "lone-load->vmcnt(0) pairs"}' "$OUT"
