"""Kernel & engine performance probe on one MI355X.

Times the hot kernels (bandwidth/TFLOPs) and the engine-level rates
(encoder texts/s, generator prefill tok/s + decode tok/s, cosine scan GB/s)
to anchor bench.py's configuration.  Writes JSON to gpurun_out/perf.json.
"""

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from sentio_amd import ops

DEV = "cuda:0"
RESULTS = {}


def timeit(fn, warmup=3, iters=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def bench_rmsnorm():
    x = torch.randn(8192, 4096, dtype=torch.bfloat16, device=DEV)
    w = torch.randn(4096, dtype=torch.bfloat16, device=DEV)
    t = timeit(lambda: ops.rmsnorm(x, w))
    byte = x.numel() * 2 * 2  # read + write
    RESULTS["rmsnorm_tb_s"] = byte / t / 1e12


def bench_swiglu():
    g = torch.randn(8192, 14336, dtype=torch.bfloat16, device=DEV)
    u = torch.randn_like(g)
    t = timeit(lambda: ops.swiglu(g, u))
    RESULTS["swiglu_tb_s"] = g.numel() * 2 * 3 / t / 1e12


def bench_gemm():
    for n in (4096, 8192):
        a = torch.randn(n, n, dtype=torch.bfloat16, device=DEV)
        b = torch.randn(n, n, dtype=torch.bfloat16, device=DEV)
        t = timeit(lambda: ops.gemm_bf16(a, b), warmup=2, iters=5)
        RESULTS[f"gemm_hand_tf_{n}"] = 2 * n**3 / t / 1e12
        t = timeit(lambda: a @ b, warmup=2, iters=5)
        RESULTS[f"gemm_rocblas_tf_{n}"] = 2 * n**3 / t / 1e12


def bench_flash():
    B, S, H, Hkv, D = 8, 2048, 32, 8, 128
    q = torch.randn(B, S, H, D, dtype=torch.bfloat16, device=DEV)
    k = torch.randn(B, S, Hkv, D, dtype=torch.bfloat16, device=DEV)
    v = torch.randn(B, S, Hkv, D, dtype=torch.bfloat16, device=DEV)
    t = timeit(lambda: ops.attention(q, k, v, causal=True), warmup=2, iters=5)
    flops = 4.0 * B * H * S * S * D * 0.5  # causal half
    RESULTS["flash_attn_tf"] = flops / t / 1e12
    RESULTS["flash_attn_ms"] = t * 1e3


def bench_decode_attn():
    B, H, Hkv, Smax, D = 16, 32, 8, 4096, 128
    q = torch.randn(B, H, D, dtype=torch.bfloat16, device=DEV)
    kc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=DEV)
    vc = torch.randn_like(kc)
    lens = torch.full((B,), Smax, dtype=torch.int32, device=DEV)
    t = timeit(lambda: ops.decode_attention(q, kc, vc, lens))
    byte = 2 * B * Hkv * Smax * D * 2
    RESULTS["decode_attn_tb_s"] = byte / t / 1e12
    RESULTS["decode_attn_us"] = t * 1e6


def bench_cosine():
    N, D, B = 2_000_000, 1024, 8
    mat = torch.randn(N, D, dtype=torch.float16, device=DEV)
    q = torch.randn(B, D, dtype=torch.float16, device=DEV)
    t = timeit(lambda: ops.cosine_topk(q, mat, 64), warmup=2, iters=5)
    RESULTS["cosine_scan_tb_s"] = N * D * 2 / t / 1e12
    RESULTS["cosine_scan_ms_2M"] = t * 1e3


def bench_encoder():
    from sentio_amd.engines.encoder import EncoderEngine

    enc = EncoderEngine("sentio-encoder-base", device=DEV, max_seq=512)
    texts = ["sample document text for embedding " * 10] * 64
    t = timeit(lambda: enc.embed(texts), warmup=2, iters=5)
    RESULTS["encoder_texts_s"] = 64 / t
    # serving/bench query shape: short sequences, and the per-request
    # single-text latency (encoder stage cost)
    encq = EncoderEngine("sentio-encoder-base", device=DEV, max_seq=128)
    queries = ["what does this mean for retrieval?"] * 32
    t = timeit(lambda: encq.embed(queries), warmup=3, iters=10)
    RESULTS["encoder_query_batch32_ms"] = t * 1e3
    t = timeit(lambda: encq.embed(queries[:1]), warmup=3, iters=20)
    RESULTS["encoder_query_single_ms"] = t * 1e3


def bench_generator(name="llama3-1b", B=8, S=512, new=32, tag=None):
    from sentio_amd.engines.generator import GeneratorEngine
    from sentio_amd.engines.transformer import KVCache

    tag = tag or name
    t0 = time.time()
    g = GeneratorEngine(name, device=DEV, max_seq=S + new + 8)
    torch.cuda.synchronize()
    RESULTS[f"{tag}_init_s"] = time.time() - t0

    tokens = torch.randint(3, 258, (B, S), device=DEV)
    cache = KVCache(g.cfg, B, S + new + 4, DEV, g.model.dtype)

    def prefill():
        cache.seq_lens[:] = 0
        return g.model.prefill(tokens, cache)

    t = timeit(prefill, warmup=1, iters=3)
    RESULTS[f"{tag}_prefill_tok_s"] = B * S / t

    logits = prefill()
    cur = logits.argmax(-1, keepdim=True)

    def decode():
        return g.model.decode_step(cur, cache)

    t = timeit(decode, warmup=3, iters=10)
    RESULTS[f"{tag}_decode_tok_s"] = B / t
    RESULTS[f"{tag}_decode_step_ms"] = t * 1e3


def main():
    if not torch.cuda.is_available():
        print(f"{__file__}: needs a GPU (MI355X) — skipping")
        return

    os.makedirs("gpurun_out", exist_ok=True)
    which = sys.argv[1] if len(sys.argv) > 1 else "all"
    steps = [
        ("rmsnorm", bench_rmsnorm), ("swiglu", bench_swiglu),
        ("gemm", bench_gemm), ("flash", bench_flash),
        ("decode_attn", bench_decode_attn), ("cosine", bench_cosine),
        ("encoder", bench_encoder),
        ("gen1b", lambda: bench_generator("llama3-1b")),
        ("gen8b", lambda: bench_generator("llama3-8b", B=8, S=512)),
        # flagship decode anchor: batch 32, ~900-token prompts (BPE era)
        ("gen8b32", lambda: bench_generator("llama3-8b", B=32, S=896,
                                            new=128, tag="llama3-8b_b32")),
    ]
    for name, fn in steps:
        if which != "all" and which != name:
            continue
        try:
            fn()
            print(name, "ok", flush=True)
        except Exception as e:
            RESULTS[name + "_error"] = str(e)
            print(name, "ERROR", e, flush=True)
    print(json.dumps(RESULTS, indent=1))
    with open("gpurun_out/perf.json", "w") as f:
        json.dump(RESULTS, f, indent=1)


if __name__ == "__main__":
    main()
