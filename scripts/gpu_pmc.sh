#!/bin/bash
set -x
export TMPDIR=/tmp
cd /tmp
PYTHONPATH="$GRAFT_REPO_ROOT" timeout 400 rocprofv3 --pmc FETCH_SIZE,WRITE_SIZE,SQ_INSTS_VALU_MFMA_MOPS_BF16,SQ_LDS_BANK_CONFLICT,SQ_BUSY_CYCLES --output-format csv -d /tmp/pmc -o p -- python "$GRAFT_REPO_ROOT/bench.py" --steps 1 --warmup 1 --docs-per-gpu 300000 --batch 16 --gen-tokens 48 > /tmp/pmc_run.log 2>&1
echo "pmc rc=$?"; tail -2 /tmp/pmc_run.log
python - <<'PYEOF'
import csv, glob, collections
agg = collections.defaultdict(lambda: collections.defaultdict(float))
calls = collections.defaultdict(int)
for f in glob.glob('/tmp/pmc/*counter*.csv'):
    for r in csv.DictReader(open(f)):
        kn = r.get('Kernel_Name', '')
        short = kn.split('(')[0].split('<')[0][:48]
        agg[short][r['Counter_Name']] += float(r['Counter_Value'])
        if r['Counter_Name'] == 'FETCH_SIZE':
            calls[short] += 1
import json
out = {}
for k, v in sorted(agg.items(), key=lambda kv: -kv[1].get('SQ_BUSY_CYCLES', 0))[:12]:
    out[k] = {c: f"{x:.3e}" for c, x in v.items()}
    out[k]['dispatches'] = calls[k]
print(json.dumps(out, indent=1))
PYEOF
