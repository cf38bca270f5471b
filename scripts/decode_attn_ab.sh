#!/bin/bash
# A/B decode-attention kernel variants on one box
for v in "$@"; do
  cp variants/hip_$v.so sentio_amd/ops/_sentio_hip.so  # variants/ is scratch: stage .so builds there before calling
  echo "=== variant $v ==="
  python - <<'PY'
import os, time, torch, sys
sys.path.insert(0, '.')
from sentio_amd import ops
dev = "cuda:0"
def bench(q, kc, vc, lens, iters=200):
    for _ in range(30): ops.decode_attention(q, kc, vc, lens)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): ops.decode_attention(q, kc, vc, lens)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters
for B, slen in [(16, 1600), (32, 1600), (32, 512)]:
    H, Hkv, Smax, D = 32, 8, 2120, 128
    torch.manual_seed(0)
    q = torch.randn(B, H, D, dtype=torch.bfloat16, device=dev)
    kc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=dev)
    vc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=dev)
    lens = torch.full((B,), slen, dtype=torch.int32, device=dev)
    kv = 2*B*Hkv*slen*D*2
    best = (1e9, None)
    for splits in (1,2,4):
        os.environ["SENTIO_DECODE_SPLITS"] = str(splits)
        t = bench(q, kc, vc, lens)
        print(f"B={B} slen={slen} splits={splits}: {t*1e6:7.1f}us {kv/t/1e12:.2f} TB/s")
    # numerics check vs fp32 ref
    os.environ.pop("SENTIO_DECODE_SPLITS", None)
    out = ops.decode_attention(q, kc, vc, lens)
    ref = ops.torch_ref.decode_attention(q.float().cpu(), kc.float().cpu(), vc.float().cpu(), lens.cpu())
    err = (out.cpu().float() - ref).abs().max().item()
    print(f"  max_err={err:.4f}")
PY
done
