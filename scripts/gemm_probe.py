"""Probe decode-shaped skinny GEMM layouts on MI355X.

Decode GEMMs at batch B read the whole weight once per token — they are
weight-bandwidth-bound.  Compares: (a) x @ W  ([K,N] row-major, the current
path), (b) F.linear(x, Wt) ([N,K] "TN" layout), (c) x @ Wt.t() view,
for the llama3-8b per-layer shapes + lm_head.  Prints achieved TB/s.
Run: gpurun -- 'python scripts/gemm_probe.py'
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

SHAPES = [  # (name, K, N) llama3-8b decode
    ("wqkv", 4096, 6144),
    ("wo", 4096, 4096),
    ("gate_up", 4096, 28672),
    ("down", 14336, 4096),
    ("lm_head", 4096, 128256),
]


def bench(fn, iters=50):
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    assert torch.cuda.is_available()
    dev = "cuda:0"
    for B in (16, 32):
        print(f"=== batch {B} ===")
        for name, K, N in SHAPES:
            x = torch.randn(B, K, dtype=torch.bfloat16, device=dev)
            w = torch.randn(K, N, dtype=torch.bfloat16, device=dev)
            wt = w.t().contiguous()          # [N, K]
            bytes_w = K * N * 2

            t_nn = bench(lambda: x @ w)
            t_lin = bench(lambda: torch.nn.functional.linear(x, wt))
            t_tv = bench(lambda: x @ wt.t())
            best = min(t_nn, t_lin, t_tv)
            print(f"{name:8s} K={K:6d} N={N:6d}  "
                  f"x@W {t_nn*1e6:7.1f}us ({bytes_w/t_nn/1e12:.2f} TB/s)  "
                  f"linear {t_lin*1e6:7.1f}us ({bytes_w/t_lin/1e12:.2f} TB/s)  "
                  f"x@Wt.t() {t_tv*1e6:7.1f}us ({bytes_w/t_tv/1e12:.2f} TB/s)")


def skinny_probe():
    from sentio_amd import ops
    dev = "cuda:0"
    print("=== skinny_gemm vs F.linear (W [N,K]) ===")
    for B in (16, 32):
        for name, K, N in SHAPES:
            x = torch.randn(B, K, dtype=torch.bfloat16, device=dev)
            wt = torch.randn(N, K, dtype=torch.bfloat16, device=dev)
            bytes_w = K * N * 2
            t_lin = bench(lambda: torch.nn.functional.linear(x, wt))
            t_sk = bench(lambda: ops.skinny_gemm(x, wt))
            print(f"B={B} {name:8s} linear {t_lin*1e6:7.1f}us "
                  f"({bytes_w/t_lin/1e12:.2f} TB/s)  skinny {t_sk*1e6:7.1f}us "
                  f"({bytes_w/t_sk/1e12:.2f} TB/s)")


if __name__ == "__main__":
    main()
    skinny_probe()
