"""Tiny real engines on CPU: encoder/reranker/generator forward paths run the
actual transformer code (fp32 on CPU) — the same code path the GPU runs with
HIP kernels."""

import torch

from sentio_amd.engines.encoder import EncoderEngine
from sentio_amd.engines.generator import GeneratorEngine
from sentio_amd.engines.reranker import RerankerEngine
from sentio_amd.models.document import Document


def test_encoder_shapes_and_determinism():
    enc = EncoderEngine("tiny-encoder", device="cpu", max_seq=64)
    out = enc.embed(["hello world", "another text"])
    assert out.shape == (2, 64)
    torch.testing.assert_close(out.norm(dim=1), torch.ones(2), rtol=1e-4, atol=1e-4)
    out2 = enc.embed(["hello world", "another text"])
    torch.testing.assert_close(out, out2)
    # different texts give different vectors
    assert not torch.allclose(out[0], out[1])


def test_encoder_batch_invariance():
    enc = EncoderEngine("tiny-encoder", device="cpu", max_seq=64)
    a = enc.embed(["same text", "padding partner that is longer"])[0]
    b = enc.embed(["same text"])[0]
    torch.testing.assert_close(a, b, rtol=1e-4, atol=1e-4)


def test_reranker_scores_and_truncation():
    rr = RerankerEngine("tiny-reranker", device="cpu", max_seq=64)
    docs = [Document(text=f"doc {i}", id=str(i)) for i in range(6)]
    out = rr.rerank("query", docs, top_k=2)
    assert len(out) == 2
    assert all(0.0 <= d.metadata["rerank_score"] <= 1.0 for d in out)
    s = out[0].metadata["rerank_score"]
    assert s >= out[1].metadata["rerank_score"]


def test_generator_greedy_deterministic_and_streams():
    g = GeneratorEngine("tiny-decoder", device="cpu", max_seq=128)
    out1 = g.generate(["prompt"], max_new_tokens=8, temperature=0.0)
    out2 = g.generate(["prompt"], max_new_tokens=8, temperature=0.0)
    assert out1 == out2
    assert isinstance(out1[0], str)
    deltas = list(g.stream("prompt", max_new_tokens=8, temperature=0.0))
    assert "".join(deltas) == out1[0]


def test_generator_batch():
    g = GeneratorEngine("tiny-decoder", device="cpu", max_seq=128)
    outs = g.generate(["a", "bb", "ccc"], max_new_tokens=4, temperature=0.0)
    assert len(outs) == 3


def test_decode_matches_prefill_consistency():
    """The incremental decode path must agree with full-prefill logits."""
    g = GeneratorEngine("tiny-decoder", device="cpu", max_seq=64)
    from sentio_amd.engines.transformer import KVCache

    ids = g.tokenizer.encode("consistency test", 32)
    tokens = torch.tensor([ids])
    cache = KVCache(g.cfg, 1, 64, "cpu", g.model.dtype)
    logits_prefill = g.model.prefill(tokens, cache)

    # now replay: prefill n-1 tokens, then decode the nth
    cache2 = KVCache(g.cfg, 1, 64, "cpu", g.model.dtype)
    g.model.prefill(tokens[:, :-1], cache2)
    logits_decode = g.model.decode_step(tokens[:, -1:], cache2)
    torch.testing.assert_close(logits_prefill, logits_decode, rtol=1e-3, atol=1e-3)


def test_encoder_embedding_cache():
    from sentio_amd.engines.encoder import EncoderEngine

    e = EncoderEngine("tiny-encoder", device="cpu", cache_size=16)
    v1 = e.embed(["alpha", "beta"])
    assert e.cache.hits == 0 and e.cache.misses >= 2
    v2 = e.embed(["alpha", "gamma", "beta"])
    assert e.cache.hits == 2
    import torch
    assert torch.allclose(v1[0], v2[0], atol=1e-6)
    assert torch.allclose(v1[1], v2[2], atol=1e-6)


def test_prefix_suffix_prefill_matches_full():
    """prefill_suffix over a prefix-primed cache must reproduce the plain
    full-prompt prefill logits (CPU, fp32)."""
    import torch

    from sentio_amd.engines.configs import MODEL_CONFIGS
    from sentio_amd.engines.transformer import KVCache, Transformer

    cfg = MODEL_CONFIGS["tiny-decoder64"]
    m = Transformer(cfg, device="cpu", seed=21)
    B, P, S = 2, 10, 7
    torch.manual_seed(3)
    full = torch.randint(0, cfg.vocab_size, (B, P + S))
    full[:, :P] = full[0, :P]          # shared prefix

    c_full = KVCache(cfg, B, 64, "cpu", m.dtype)
    want = m.prefill(full, c_full)

    # prefix KV computed batch-1, broadcast into a fresh cache
    c_pre = KVCache(cfg, 1, P, "cpu", m.dtype)
    m.forward_hidden(full[:1, :P], cache=c_pre)
    c_suf = KVCache(cfg, B, 64, "cpu", m.dtype)
    for i in range(cfg.n_layers):
        c_suf.k[i][:, :, :P] = c_pre.k[i]
        c_suf.v[i][:, :, :P] = c_pre.v[i]
    got = m.prefill_suffix(full[:, P:], c_suf, P)
    assert torch.allclose(got, want, rtol=1e-4, atol=1e-4)
    # decode continues identically from both caches
    nxt = torch.randint(0, cfg.vocab_size, (B, 1))
    d1 = m.decode_step(nxt, c_full)
    d2 = m.decode_step(nxt, c_suf)
    assert torch.allclose(d1, d2, rtol=1e-4, atol=1e-4)


def test_generate_with_prefix_kv_matches_disabled(monkeypatch):
    """Full generate() path: identical greedy tokens with prefix caching on
    and off (CPU fp32 — exact)."""
    from sentio_amd.engines.generator import GeneratorEngine

    eng = GeneratorEngine("tiny-decoder64", device="cpu", max_seq=256)
    prefix = "system preamble shared across requests. " * 10
    prompts = [prefix + f"question number {i}?" for i in range(3)]
    monkeypatch.setenv("SENTIO_PREFIX_KV", "0")
    base = eng.generate(prompts, max_new_tokens=8, temperature=0.0,
                        stop_on_eos=False)
    monkeypatch.setenv("SENTIO_PREFIX_KV", "1")
    cached = eng.generate(prompts, max_new_tokens=8, temperature=0.0,
                          stop_on_eos=False)
    assert base == cached
    assert eng._prefix_store  # the prefix KV was actually computed/cached


def test_prefix_split_token_exactness():
    """prefix_ids + suffix ids (no BOS) must reproduce the full prompt's
    token stream exactly — regression for an off-by-one + stray-BOS split."""
    from sentio_amd.engines.generator import GeneratorEngine

    eng = GeneratorEngine("tiny-decoder64", device="cpu", max_seq=256)
    prefix = "shared preamble text. " * 16
    prompts = [prefix + f"tail {i}" for i in range(3)]
    pre_ids, suffixes = eng._split_shared_prefix(prompts, 200)
    assert len(pre_ids) >= 64
    for p, suf in zip(prompts, suffixes):
        full = eng.tokenizer.encode(p, None)
        suf_ids = eng.tokenizer.encode(suf, None, add_bos=False)
        assert pre_ids + suf_ids == full


def test_prefix_split_nonascii_token_budget():
    """The prefix cap is a TOKEN budget: multi-byte UTF-8 prefixes must not
    overshoot it (a char-based cap yielded 548 prefix tokens from a 200-token
    budget and crashed prefill — ADVICE r1 high)."""
    from sentio_amd.engines.generator import GeneratorEngine

    eng = GeneratorEngine("tiny-decoder64", device="cpu", max_seq=256)
    prefix = "共有システムプリアンブル。質問に答えてください。" * 8   # 3 B/char
    prompts = [prefix + f"質問{i}?" for i in range(3)]
    budget = 200
    pre_ids, suffixes = eng._split_shared_prefix(prompts, budget)
    assert len(pre_ids) <= budget - 8
    if pre_ids:   # BPE declines whitespace-free prefixes (no safe boundary)
        for p, suf in zip(prompts, suffixes):
            full = eng.tokenizer.encode(p, None)
            suf_ids = eng.tokenizer.encode(suf, None, add_bos=False)
            assert pre_ids + suf_ids == full
            assert len(suf_ids) >= 1
    # and the full generate() path survives a CJK shared prefix
    out = eng.generate(prompts, max_new_tokens=4, temperature=0.0,
                       stop_on_eos=False)
    assert len(out) == 3


def test_prefix_split_nonascii_byte_tokenizer(monkeypatch):
    """Byte tokenizer (fallback): CJK prefixes DO split, exactly, within the
    token budget — the original ADVICE r1 regression case."""
    monkeypatch.setenv("SENTIO_TOKENIZER", "byte")
    from sentio_amd.engines.generator import GeneratorEngine

    eng = GeneratorEngine("tiny-decoder64", device="cpu", max_seq=256)
    prefix = "共有システムプリアンブル。質問に答えてください。" * 8
    prompts = [prefix + f"質問{i}?" for i in range(3)]
    budget = 200
    pre_ids, suffixes = eng._split_shared_prefix(prompts, budget)
    assert 0 < len(pre_ids) <= budget - 8
    for p, suf in zip(prompts, suffixes):
        full = eng.tokenizer.encode(p, None)
        suf_ids = eng.tokenizer.encode(suf, None, add_bos=False)
        assert pre_ids + suf_ids == full
        assert len(suf_ids) >= 1
    out = eng.generate(prompts, max_new_tokens=4, temperature=0.0,
                       stop_on_eos=False)
    assert len(out) == 3


def test_generate_batch_composition_invariance():
    """A request's greedy output must not depend on its co-batched partners'
    prompt lengths: logits are gathered at each row's true last position and
    pad keys are masked."""
    g = GeneratorEngine("tiny-decoder64", device="cpu", max_seq=256)
    prompts = ["tell me about graphs",
               "a considerably longer prompt about distributed retrieval "
               "engines and how their sharded indexes merge candidates",
               "hi"]
    batched = g.generate(prompts, max_new_tokens=6, temperature=0.0,
                         stop_on_eos=False)
    singles = [g.generate([p], max_new_tokens=6, temperature=0.0,
                          stop_on_eos=False)[0] for p in prompts]
    assert batched == singles


def test_model_config_param_counts():
    """Config shapes actually correspond to the named model classes."""
    from sentio_amd.engines.configs import MODEL_CONFIGS

    def b(name):
        return MODEL_CONFIGS[name].n_params / 1e9

    assert 7.0 <= b("llama3-8b") <= 9.0
    assert 65 <= b("llama3-70b") <= 75
    assert 6.5 <= b("qwen2-7b") <= 8.5
    assert 6.5 <= b("mistral-7b") <= 8.0
    enc = MODEL_CONFIGS["sentio-encoder-base"]
    assert enc.dim == 1024 and not enc.causal    # jina-v3 class output dim
    rr = MODEL_CONFIGS["sentio-reranker-base"]
    assert rr.pooled_head == 1                   # scalar relevance head


def test_encoder_stats_counters():
    from sentio_amd.engines.encoder import EncoderEngine

    enc = EncoderEngine("sentio-encoder-small", device="cpu", dtype="fp32",
                        max_seq=32, cache_size=16)
    enc.embed(["alpha", "beta"])
    enc.embed(["alpha", "gamma"])      # alpha hits the cache
    s = enc.stats()
    assert s["calls"] == 2
    assert s["texts_embedded"] == 3    # alpha embedded once
    assert s["cache_hits"] == 1
    assert s["errors"] == 0
    assert s["avg_time_ms_per_text"] > 0
