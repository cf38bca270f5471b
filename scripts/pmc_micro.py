"""Tiny kernel driver for PMC collection: a few dispatches each of the
flagship hand kernels (decode attention, flash attention, cosine scan).
Run under rocprofv3 --pmc; see profiles/pmc_summary.md."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from sentio_amd import ops


def main():
    if not torch.cuda.is_available():
        print("needs GPU")
        return
    dev = "cuda:0"
    B, H, Hkv, Smax, D, slen = 16, 32, 8, 2120, 128, 1600
    q = torch.randn(B, H, D, dtype=torch.bfloat16, device=dev)
    kc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=dev)
    vc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=dev)
    lens = torch.full((B,), slen, dtype=torch.int32, device=dev)
    for _ in range(10):
        ops.decode_attention(q, kc, vc, lens)
    qf = torch.randn(4, 2048, H, D, dtype=torch.bfloat16, device=dev)
    kf = torch.randn(4, 2048, Hkv, D, dtype=torch.bfloat16, device=dev)
    vf = torch.randn(4, 2048, Hkv, D, dtype=torch.bfloat16, device=dev)
    for _ in range(5):
        ops.attention(qf, kf, vf, causal=True)
    mat = torch.randn(300000, 1024, dtype=torch.float16, device=dev)
    qq = torch.randn(16, 1024, dtype=torch.float16, device=dev)
    for _ in range(5):
        ops.cosine_topk(qq, mat, 10)
    torch.cuda.synchronize()
    print("done")


if __name__ == "__main__":
    main()
