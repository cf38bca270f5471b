#!/bin/bash
set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
timeout 300 python -m pytest tests/test_ops_gpu.py tests/test_engines_cpu.py -x -q -m gpu 2>&1 | tail -3
timeout 400 python scripts/decode_attn_probe.py 2>&1 | tail -22
timeout 700 python bench.py --steps 3 --warmup 1 > gpurun_out/bench_flagship.json 2> gpurun_out/bench_flagship.log
echo "bench rc=$?"; cat gpurun_out/bench_flagship.json | head -c 300; echo; grep stage gpurun_out/bench_flagship.log
export TMPDIR=/tmp; cd /tmp
PYTHONPATH="$GRAFT_REPO_ROOT" timeout 300 rocprofv3 --pmc FETCH_SIZE,SQ_BUSY_CYCLES,SQ_WAIT_ANY,SQ_INSTS_VMEM,SQ_INSTS_LDS --output-format csv -d /tmp/pmc -o p -- python -c "
import torch
from sentio_amd import ops
dev='cuda:0'; B,H,Hkv,Smax,D,slen=16,32,8,2120,128,1600
q=torch.randn(B,H,D,dtype=torch.bfloat16,device=dev)
kc=torch.randn(B,Hkv,Smax,D,dtype=torch.bfloat16,device=dev)
vc=torch.randn(B,Hkv,Smax,D,dtype=torch.bfloat16,device=dev)
lens=torch.full((B,),slen,dtype=torch.int32,device=dev)
for _ in range(30): ops.decode_attention(q,kc,vc,lens)
torch.cuda.synchronize()" > /tmp/pmc_run.log 2>&1
echo "pmc rc=$?"
cp /tmp/pmc/*.csv "$GRAFT_REPO_ROOT/gpurun_out/" 2>/dev/null
python - <<'PYEOF'
import csv, glob, collections
for f in glob.glob('/tmp/pmc/*counter*.csv'):
    agg = collections.defaultdict(float); n = collections.defaultdict(int)
    for r in csv.DictReader(open(f)):
        if 'decode_attn_split' in r.get('Kernel_Name',''):
            agg[r['Counter_Name']] += float(r['Counter_Value']); n[r['Counter_Name']] += 1
    for k, v in agg.items():
        print(f"{k}: total {v:.3e} over {n[k]} dispatches")
PYEOF
echo done
