"""Probe mfma_f32_16x16x32_bf16 fragment layout via the gemm_bf16 kernel.

Runs a tiny GEMM with basis-vector operands and prints where mass lands —
diagnoses row/col swaps in the assumed lane→element maps (guide §3: always
check with asymmetric operands).
"""

import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from sentio_amd import ops


def main():
    if not torch.cuda.is_available():
        print(f"{__file__}: needs a GPU (MI355X) — skipping")
        return

    dev = "cuda:0"
    M = K = N = 128
    a = torch.zeros(M, K, dtype=torch.bfloat16, device=dev)
    b = torch.zeros(K, N, dtype=torch.bfloat16, device=dev)
    # asymmetric pattern: A[i][k] = delta(i==2, k==5); B[k][j] = delta(k==5, j==9)
    a[2, 5] = 1.0
    b[5, 9] = 1.0
    c = ops.gemm_bf16(a, b)
    nz = c.nonzero()
    print("single-element probe nonzeros:", nz.cpu().tolist()[:5],
          "value:", c[2, 9].item() if nz.numel() else None)

    # full random check
    torch.manual_seed(0)
    a = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
    b = torch.randn(K, N, dtype=torch.bfloat16, device=dev)
    c = ops.gemm_bf16(a, b)
    want = (a.float() @ b.float())
    err = (c.float() - want).abs().max().item()
    rel = err / want.abs().max().item()
    print(f"random 128^3 max abs err {err:.4f} rel {rel:.5f}")

    # attention probe: single head small
    B, S, H, D = 1, 32, 1, 64
    q = torch.randn(B, S, H, D, dtype=torch.bfloat16, device=dev)
    k = torch.randn(B, S, H, D, dtype=torch.bfloat16, device=dev)
    v = torch.randn(B, S, H, D, dtype=torch.bfloat16, device=dev)
    got = ops.attention(q, k, v, causal=True)
    want = ops.torch_ref.attention(q.cpu().float(), k.cpu().float(),
                                   v.cpu().float(), causal=True)
    err = (got.float().cpu() - want).abs().max().item()
    print(f"attention 32x64 max abs err {err:.4f}")


if __name__ == "__main__":
    main()
