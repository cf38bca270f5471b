"""Isolated decode-attention microbenchmark (llama3-8b decode shape).
Prints achieved µs + effective KV TB/s per splits setting.
Run: gpurun -- 'python scripts/decode_attn_probe.py'"""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from sentio_amd import ops

def bench(q, kc, vc, lens, iters=200):
    for _ in range(20):
        ops.decode_attention(q, kc, vc, lens)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        ops.decode_attention(q, kc, vc, lens)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters

def main():
    if not torch.cuda.is_available():
        print(f"{__file__}: needs a GPU (MI355X) — skipping")
        return

    dev = "cuda:0"
    for B, slen in [(16, 1600), (32, 1600), (32, 512)]:
        H, Hkv, Smax, D = 32, 8, 2120, 128
        torch.manual_seed(0)
        q = torch.randn(B, H, D, dtype=torch.bfloat16, device=dev)
        kc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=dev)
        vc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=dev)
        lens = torch.full((B,), slen, dtype=torch.int32, device=dev)
        kv_bytes = 2 * B * Hkv * slen * D * 2
        print(f"--- B={B} slen={slen} (KV {kv_bytes/1e6:.0f} MB)")
        for splits in (0, 1, 2, 4, 8):
            if splits: os.environ["SENTIO_DECODE_SPLITS"] = str(splits)
            else: os.environ.pop("SENTIO_DECODE_SPLITS", None)
            t = bench(q, kc, vc, lens)
            print(f"splits={splits or 'auto'}: {t*1e6:7.1f}us  {kv_bytes/t/1e12:.2f} TB/s")




def bmm_probe():
    """Compare the hand decode-attention kernel against a batched-GEMM
    formulation (rocBLAS bmm): scores = Q·K^T and O = P·V per (b, kv-head)
    group, full-Smax with masking."""
    dev = "cuda:0"
    for B, slen in [(16, 1600), (32, 1600)]:
        H, Hkv, Smax, D = 32, 8, 2120, 128
        G = H // Hkv
        torch.manual_seed(0)
        q = torch.randn(B, H, D, dtype=torch.bfloat16, device=dev)
        kc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=dev)
        vc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=dev)
        lens = torch.full((B,), slen, dtype=torch.int32, device=dev)
        mask = (torch.arange(Smax, device=dev)[None, :]
                >= lens[:, None]).view(B, 1, 1, Smax)

        def bmm_path():
            qg = q.view(B * Hkv, G, D)
            kg = kc.view(B * Hkv, Smax, D)
            s_ = torch.bmm(qg, kg.transpose(1, 2)) * (D ** -0.5)
            s_ = s_.view(B, Hkv, G, Smax).masked_fill(mask, float("-inf"))
            p_ = torch.softmax(s_.float(), dim=-1).to(torch.bfloat16)
            return torch.bmm(p_.view(B * Hkv, G, Smax),
                             vc.view(B * Hkv, Smax, D))

        for name, fn in [("hand", lambda: ops.decode_attention(q, kc, vc, lens)),
                         ("bmm", bmm_path)]:
            for _ in range(10):
                fn()
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(100):
                fn()
            torch.cuda.synchronize()
            t = (time.perf_counter() - t0) / 100
            kv_bytes = 2 * B * Hkv * slen * D * 2
            print(f"B={B} {name}: {t*1e6:7.1f}us  {kv_bytes/t/1e12:.2f} TB/s")


def flash_probe():
    """Prefill flash-attention microbench (llama3-8b shape)."""
    dev = "cuda:0"
    for B, S in [(16, 2048), (32, 2048)]:
        H, Hkv, D = 32, 8, 128
        q = torch.randn(B, S, H, D, dtype=torch.bfloat16, device=dev)
        k = torch.randn(B, S, Hkv, D, dtype=torch.bfloat16, device=dev)
        v = torch.randn(B, S, Hkv, D, dtype=torch.bfloat16, device=dev)
        for _ in range(5):
            ops.attention(q, k, v, causal=True)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        iters = 20
        for _ in range(iters):
            ops.attention(q, k, v, causal=True)
        torch.cuda.synchronize()
        t = (time.perf_counter() - t0) / iters
        flops = 2.0 * B * H * S * S * D * 2 / 2  # causal half
        print(f"flash B={B} S={S}: {t*1e3:.2f} ms  {flops/t/1e12:.0f} TFLOP/s")


if __name__ == "__main__":
    main()
    bmm_probe()
    flash_probe()
