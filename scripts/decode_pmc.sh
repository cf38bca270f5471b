#!/bin/bash
# PMC evidence for the decode-attention kernel in isolation:
# pass 1: wave-state split (parked vs issue-stall vs active)
# pass 2: memory-side bytes + L2 hit rate
set -x
export TMPDIR=/tmp
cd /tmp
cat > /tmp/dec_iso.py <<'PY'
import os, sys, torch
sys.path.insert(0, os.environ["GRAFT_REPO_ROOT"])
from sentio_amd import ops
dev = "cuda:0"
B, slen, H, Hkv, Smax, D = 32, 1600, 32, 8, 2120, 128
os.environ["SENTIO_DECODE_SPLITS"] = os.environ.get("SPLITS", "2")
torch.manual_seed(0)
q = torch.randn(B, H, D, dtype=torch.bfloat16, device=dev)
kc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=dev)
vc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=dev)
lens = torch.full((B,), slen, dtype=torch.int32, device=dev)
for _ in range(50):
    ops.decode_attention(q, kc, vc, lens)
torch.cuda.synchronize()
PY
for SET in "SQ_WAIT_ANY,SQ_WAIT_INST_ANY,SQ_ACTIVE_INST_ANY,SQ_WAVE_CYCLES,SQ_BUSY_CYCLES" "FETCH_SIZE,TCC_HIT_sum,TCC_MISS_sum" "WRITE_SIZE,SQ_LDS_BANK_CONFLICT"; do
  rm -rf /tmp/pmcdec; timeout 300 rocprofv3 --pmc "$SET" --output-format csv -d /tmp/pmcdec -o p -- python /tmp/dec_iso.py > /tmp/dec_pmc.log 2>&1
  echo "rc=$? set=$SET"
  python - <<'PYEOF'
import csv, glob, collections
agg = collections.defaultdict(lambda: collections.defaultdict(float))
n = collections.defaultdict(lambda: collections.defaultdict(int))
for f in glob.glob('/tmp/pmcdec/*counter*.csv'):
    for r in csv.DictReader(open(f)):
        kn = r.get('Kernel_Name', '').split('(')[0].split('<')[0][:40]
        agg[kn][r['Counter_Name']] += float(r['Counter_Value'])
        n[kn][r['Counter_Name']] += 1
for k, v in agg.items():
    if 'decode' not in k and 'combine' not in k:
        continue
    print(k, {c: f"{x/max(n[k][c],1):.4g}" for c, x in v.items()}, "dispatches", max(n[k].values()))
PYEOF
done
