import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from sentio_amd import ops

dev = "cuda:0"
for B, H, Hkv in [(2, 2, 1), (1, 8, 2), (1, 8, 8), (2, 8, 2)]:
    S, Smax, D, P = 40, 128, 128, 30
    torch.manual_seed(12)
    q = torch.randn(B, S, H, D, dtype=torch.bfloat16, device=dev)
    kc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=dev)
    vc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=dev)
    lens = torch.full((B,), P + S, dtype=torch.int32, device=dev)
    got = ops.attention_cache(q, kc, vc, lens, P).float().cpu()
    want = ops.torch_ref.attention_cache(q.cpu().float(), kc.cpu().float(),
                                         vc.cpu().float(), lens.cpu(), P).float()
    err = (got - want).abs()
    print(f"B={B} H={H} Hkv={Hkv}: max err {err.max():.4f} "
          f"worst head {err.amax(dim=(0,1,3)).argmax().item()}")
