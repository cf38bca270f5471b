#!/bin/bash
# GPU serving load test: real engines + uvicorn +
# dynamic batcher on 1xMI355X, mixed stream + non-stream traffic at 16 and
# 32 connections.  Run under gpurun; results land in gpurun_out/ and the
# curated copy goes to profiles/.  Every stage is bounded by `timeout` so a
# server wedge can never hang the box.
set -x
export DISABLE_AUTH=true
export RATE_LIMIT_CHAT_PER_MIN=1000000
export RATE_LIMIT_EMBED_PER_MIN=1000000
export LLM_MAX_TOKENS=64
export USE_VERIFIER=false
mkdir -p gpurun_out
python -m sentio_amd.serving.app > gpurun_out/server.log 2>&1 &
SRV=$!
# wait for the engines to come up (model init + bucket graph captures)
for i in $(seq 1 150); do
  sleep 2
  curl -sf -m 5 http://127.0.0.1:8000/health > /dev/null && break
done
curl -sf -m 5 http://127.0.0.1:8000/health || { tail -20 gpurun_out/server.log; kill $SRV; exit 1; }
# seed a small corpus through /embed, then a warmup burst
timeout 300 python scripts/load_test.py --clients 4 --requests 12 --seed-docs 40 \
  --stream-frac 0.25 > gpurun_out/load_warmup.json 2>gpurun_out/load_warmup.err
# measured runs: 16 and 32 connections, 25% SSE streams
timeout 420 python scripts/load_test.py --clients 16 --requests 128 \
  --stream-frac 0.25 > gpurun_out/load_test_gpu_16c128r.json
cat gpurun_out/load_test_gpu_16c128r.json
timeout 420 python scripts/load_test.py --clients 32 --requests 192 \
  --stream-frac 0.25 > gpurun_out/load_test_gpu_32c192r.json
cat gpurun_out/load_test_gpu_32c192r.json
timeout 420 python scripts/load_test.py --clients 48 --requests 240 \
  --stream-frac 0.25 > gpurun_out/load_test_gpu_48c240r.json
cat gpurun_out/load_test_gpu_48c240r.json
# batcher coalescing evidence + perf counters
curl -s -m 30 http://127.0.0.1:8000/health/detailed > gpurun_out/load_health_detailed.json
curl -s -m 30 http://127.0.0.1:8000/metrics/performance > gpurun_out/load_metrics_perf.json
kill $SRV
wait $SRV 2>/dev/null
exit 0
