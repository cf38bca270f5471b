"""The shared transformer core for encoder / reranker / generator engines.

MI355X-first design notes:
* plain projection GEMMs (fused QKV, fused gate+up, down, lm_head) go through
  torch.matmul → hipBLASLt/rocBLAS on ROCm — library GEMMs for library-shaped
  work;
* everything between them is a hand-written gfx950 HIP kernel via sentio_amd.ops:
  fused RMSNorm (+residual), RoPE from precomputed tables, flash-style MFMA
  attention for prefill, single-pass online-softmax decode attention over the
  KV cache, fused SwiGLU, masked mean-pool + L2-norm for the encoder;
* KV caches are preallocated contiguous [B, Hkv, Smax, hd] tensors in HBM —
  288 GB/GPU means we size for residency, not paging;
* weights are bf16, random-init (no network for checkpoints).

Replaces (semantically) the reference's remote encoder/reranker/LLM calls
(reference src/core/embeddings/providers/jina.py:165,
src/core/rerankers/jina_reranker.py:172, src/core/llm/providers/openai.py:117).
"""

from __future__ import annotations

import math

import torch

from sentio_amd import ops
from sentio_amd.engines.configs import ModelConfig
from sentio_amd.parallel.tp import (TPContext, linear_row_parallel,
                                    shard_columns, shard_qkv, shard_rows)


def _dtype(name: str) -> torch.dtype:
    return {"bf16": torch.bfloat16, "fp16": torch.float16, "fp32": torch.float32}[name]


class TransformerWeights:
    """Flat weight container (not nn.Module: no autograd needed for serving)."""

    def __init__(self, cfg: ModelConfig, device: str, dtype: torch.dtype,
                 seed: int = 1234, tp: TPContext | None = None):
        self.cfg = cfg
        self.device = device
        self.dtype = dtype
        self.tp = tp or TPContext()
        gen = torch.Generator(device="cpu")
        gen.manual_seed(seed)
        d, hd = cfg.dim, cfg.head_dim
        qkv_out = (cfg.n_heads + 2 * cfg.n_kv_heads) * hd
        std = 0.02
        f = cfg.ffn_dim

        def init(*shape):
            # init on device when possible for speed; generator is CPU-side so
            # large models draw on device with a per-tensor seed instead.
            if device == "cpu":
                return torch.randn(*shape, generator=gen, dtype=torch.float32).mul_(std).to(dtype)
            t = torch.empty(*shape, dtype=dtype, device=device)
            t.normal_(0.0, std)
            return t

        if device != "cpu":
            torch.manual_seed(seed)

        tp = self.tp
        self.tok_emb = init(cfg.vocab_size, d)
        self.layers = []
        # Weights are STORED [out, in] (TN layout) and applied via F.linear:
        # hipBLASLt's TN path streams the weight rows coalesced for skinny
        # decode batches (measured 2.5x on the down projection vs [in, out],
        # profiles/gemm_probe) and is the layout checkpoints use.
        for _ in range(cfg.n_layers):
            # full tensors are drawn identically on every rank (same seed and
            # order), then sliced — TP=N matches TP=1 numerically.
            wqkv = shard_qkv(init(d, qkv_out), cfg.n_heads, cfg.n_kv_heads, hd, tp)
            wo = shard_rows(init(cfg.n_heads * hd, d), tp)
            wgu_full = init(d, 2 * f)
            if tp.enabled:
                w_gate_up = torch.cat(
                    [shard_columns(wgu_full[:, :f], tp),
                     shard_columns(wgu_full[:, f:], tp)], dim=1)
            else:
                w_gate_up = wgu_full
            w_down = shard_rows(init(f, d), tp)
            self.layers.append({
                "attn_norm": torch.ones(d, dtype=dtype, device=device),
                "wqkv": wqkv.t().contiguous(),
                "wo": wo.t().contiguous(),
                "ffn_norm": torch.ones(d, dtype=dtype, device=device),
                "w_gate_up": w_gate_up.t().contiguous(),
                "w_down": w_down.t().contiguous(),
            })
        self.final_norm = torch.ones(d, dtype=dtype, device=device)
        if cfg.causal:
            self.lm_head = init(d, cfg.vocab_size).t().contiguous()
        if cfg.pooled_head:
            self.head = init(d, cfg.pooled_head).t().contiguous()

    @property
    def n_bytes(self) -> int:
        total = self.tok_emb.nelement()
        for l in self.layers:
            total += sum(t.nelement() for t in l.values())
        total += self.final_norm.nelement()
        if hasattr(self, "lm_head"):
            total += self.lm_head.nelement()
        return total * self.tok_emb.element_size()


class KVCache:
    def __init__(self, cfg: ModelConfig, batch: int, max_seq: int, device: str,
                 dtype: torch.dtype, n_kv_heads: int | None = None):
        nkv = n_kv_heads or cfg.n_kv_heads
        self.k = [
            torch.zeros(batch, nkv, max_seq, cfg.head_dim,
                        dtype=dtype, device=device)
            for _ in range(cfg.n_layers)
        ]
        self.v = [
            torch.zeros(batch, nkv, max_seq, cfg.head_dim,
                        dtype=dtype, device=device)
            for _ in range(cfg.n_layers)
        ]
        self.seq_lens = torch.zeros(batch, dtype=torch.int32, device=device)
        self.max_seq = max_seq


class Transformer:
    """Forward-only transformer executing on sentio ops."""

    def __init__(self, cfg: ModelConfig, device: str = "cpu",
                 dtype: str | torch.dtype = "bf16", seed: int = 1234,
                 tp: TPContext | None = None):
        self.cfg = cfg
        self.device = device
        self.tp = tp or TPContext()
        if self.tp.enabled:
            assert cfg.n_heads % self.tp.world == 0
            assert cfg.n_kv_heads % self.tp.world == 0
        self.h_local = cfg.n_heads // self.tp.world
        self.hkv_local = cfg.n_kv_heads // self.tp.world
        self.dtype = _dtype(dtype) if isinstance(dtype, str) else dtype
        if device == "cpu":
            # bf16 matmuls on CPU are slow and loose; tests run fp32
            self.dtype = torch.float32
        self.w = TransformerWeights(cfg, device, self.dtype, seed, self.tp)
        cos, sin = ops.torch_ref.rope_tables(cfg.max_seq, cfg.head_dim,
                                             cfg.rope_base, device)
        self.rope_cos, self.rope_sin = cos, sin
        self.scale = 1.0 / math.sqrt(cfg.head_dim)

    # ----- core blocks -----
    def _attn(self, x: torch.Tensor, layer: dict, pos: torch.Tensor,
              cache: KVCache | None, layer_idx: int,
              kv_lens: torch.Tensor | None, q_off: int = 0) -> torch.Tensor:
        cfg = self.cfg
        B, S, d = x.shape
        hd = cfg.head_dim
        H, Hkv = self.h_local, self.hkv_local
        qkv = torch.nn.functional.linear(x.view(B * S, d), layer["wqkv"])
        qkv = qkv.view(B, S, -1)
        q_end = H * hd
        k_end = q_end + Hkv * hd
        q = qkv[..., :q_end].reshape(B, S, H, hd)
        k = qkv[..., q_end:k_end].reshape(B, S, Hkv, hd)
        v = qkv[..., k_end:].reshape(B, S, Hkv, hd)
        q = ops.rope_apply(q, self.rope_cos, self.rope_sin, pos)
        k = ops.rope_apply(k, self.rope_cos, self.rope_sin, pos)

        if cache is not None:
            # write k/v at pos into the cache: [B, Hkv_local, Smax, hd]
            kc, vc = cache.k[layer_idx], cache.v[layer_idx]
            idx = pos.long()  # [B,S]
            bidx = torch.arange(B, device=x.device).unsqueeze(1).expand(B, S)
            kc[bidx.reshape(-1), :, idx.reshape(-1)] = \
                k.reshape(B * S, Hkv, hd).to(kc.dtype)
            vc[bidx.reshape(-1), :, idx.reshape(-1)] = \
                v.reshape(B * S, Hkv, hd).to(vc.dtype)

        if S == 1 and cache is not None:
            seq_lens = (pos[:, 0] + 1).to(torch.int32)
            out = ops.decode_attention(
                q.view(B, H, hd), cache.k[layer_idx],
                cache.v[layer_idx], seq_lens, self.scale,
            ).view(B, 1, H, hd)
        elif cache is not None and q_off > 0:
            # prefix-KV-cached suffix prefill: attend to cache rows
            # [0, kv_lens[b]) — the prefix KV was computed once and copied
            # in; kv_lens carries per-row ABSOLUTE lengths (prefix + true
            # suffix length), masking right-pad keys out of shorter rows
            lens_abs = (kv_lens if kv_lens is not None
                        else torch.full((B,), q_off + S, dtype=torch.int32,
                                        device=x.device))
            out = ops.attention_cache(q, cache.k[layer_idx],
                                      cache.v[layer_idx], lens_abs,
                                      q_off, self.scale)
        else:
            out = ops.attention(q, k, v, causal=cfg.causal, scale=self.scale,
                                kv_lens=kv_lens)
        # row-parallel output projection: partial-sum all-reduce overlapped
        # chunk-wise with the GEMM (parallel/tp.py)
        out = linear_row_parallel(out.reshape(B * S, H * hd), layer["wo"],
                                  self.tp)
        return out.view(B, S, d)

    def _ffn(self, x: torch.Tensor, layer: dict) -> torch.Tensor:
        B, S, d = x.shape
        # measured: hipBLASLt's TN path beats the hand skinny_gemm kernel on
        # the decode gate_up shape in context (63 vs ~87 us at B=32); decode
        # rows additionally go through the autotuned-algorithm binding
        rows = B * S
        xx = x.view(rows, d)
        if rows <= 64 and self.device != "cpu":
            gu = ops.lt_linear(xx, layer["w_gate_up"])
            y = ops.swiglu_packed(gu)   # fused [gate|up] split + silu·up
            out = linear_row_parallel(y, layer["w_down"], self.tp,
                                      linear=ops.lt_linear)
        else:
            gu = torch.nn.functional.linear(xx, layer["w_gate_up"])
            y = ops.swiglu_packed(gu)
            out = linear_row_parallel(y, layer["w_down"], self.tp)
        return out.view(B, S, d)

    def forward_hidden(
        self, tokens: torch.Tensor, pos: torch.Tensor | None = None,
        cache: KVCache | None = None, kv_lens: torch.Tensor | None = None,
        q_off: int = 0,
    ) -> torch.Tensor:
        """tokens: [B, S] int64 → hidden [B, S, dim] (after final norm).
        kv_lens: right-padding valid lengths for non-causal batches.
        Residual adds are fused into the next block's RMSNorm
        (ops.rmsnorm_residual) — one kernel instead of add + norm."""
        B, S = tokens.shape
        if pos is None:
            pos = torch.arange(S, device=tokens.device).unsqueeze(0).expand(B, S)
        x = self.w.tok_emb[tokens.reshape(-1)].view(B, S, self.cfg.dim)
        eps = self.cfg.norm_eps
        layers = self.w.layers
        n = len(layers)
        normed = ops.rmsnorm(x, layers[0]["attn_norm"], eps)
        for i, layer in enumerate(layers):
            a = self._attn(normed, layer, pos, cache, i, kv_lens, q_off=q_off)
            normed, x = ops.rmsnorm_residual(a, x, layer["ffn_norm"], eps)
            f = self._ffn(normed, layer)
            w_next = layers[i + 1]["attn_norm"] if i + 1 < n else self.w.final_norm
            normed, x = ops.rmsnorm_residual(f, x, w_next, eps)
        return normed

    # Decode-attention path choice (env override SENTIO_DECODE_ATTN=hand|bmm):
    # the bmm formulation looked ~1.4x faster in an isolated cold-box probe,
    # but IN CONTEXT the extra P-matrix traffic + per-layer kernel tail made
    # the end-to-end bench 7% slower (16.8 vs 18.1 QPS) — the hand split-S
    # kernel with fused online softmax stays the default at any fill.
    _BMM_FILL_THRESHOLD = 2.0   # auto never picks bmm

    def _decode_attn_impl(self, fill_ratio: float):
        import os

        mode = os.environ.get("SENTIO_DECODE_ATTN", "auto")
        if mode == "hand":
            return ops.decode_attention
        if mode == "bmm":
            return ops.decode_attention_bmm
        return (ops.decode_attention_bmm
                if fill_ratio >= self._BMM_FILL_THRESHOLD
                else ops.decode_attention)

    # ----- decode fast path: fused RoPE + KV-cache write + fused norms -----
    def _attn_decode(self, normed: torch.Tensor, layer: dict, cache: KVCache,
                     layer_idx: int, attn_lens: torch.Tensor,
                     attn_fn=None) -> torch.Tensor:
        B = normed.shape[0]
        d = self.cfg.dim
        hd = self.cfg.head_dim
        qkv = ops.lt_linear(normed.view(B, d), layer["wqkv"])
        q = ops.decode_qkv_prep(qkv, cache.k[layer_idx], cache.v[layer_idx],
                                self.rope_cos, self.rope_sin, cache.seq_lens)
        attn = attn_fn or ops.decode_attention
        out = attn(q, cache.k[layer_idx], cache.v[layer_idx],
                   attn_lens, self.scale)
        out = linear_row_parallel(out.view(B, self.h_local * hd),
                                  layer["wo"], self.tp, linear=ops.lt_linear)
        return out.view(B, 1, d)

    def forward_decode(self, tokens: torch.Tensor, cache: KVCache) -> torch.Tensor:
        """One-token decode: tokens [B, 1] → hidden [B, 1, dim].  Uses the
        fused decode_qkv_prep kernel (RoPE + cache write in one launch) and
        advances cache.seq_lens on device (hipGraph-capturable)."""
        B = tokens.shape[0]
        x = self.w.tok_emb[tokens.reshape(-1)].view(B, 1, self.cfg.dim)
        eps = self.cfg.norm_eps
        layers = self.w.layers
        n = len(layers)
        attn_lens = cache.seq_lens + 1
        # hand vs bmm kernel chosen at prefill time (set by the caller via
        # self.decode_attn_fn — no device sync here: this path is captured
        # into hipGraphs)
        attn_fn = getattr(self, "decode_attn_fn", None)
        normed = ops.rmsnorm(x, layers[0]["attn_norm"], eps)
        for i, layer in enumerate(layers):
            a = self._attn_decode(normed, layer, cache, i, attn_lens,
                                  attn_fn=attn_fn)
            normed, x = ops.rmsnorm_residual(a, x, layer["ffn_norm"], eps)
            f = self._ffn(normed, layer)
            w_next = layers[i + 1]["attn_norm"] if i + 1 < n else self.w.final_norm
            normed, x = ops.rmsnorm_residual(f, x, w_next, eps)
        return normed

    # ----- decoder-specific -----
    def logits(self, hidden: torch.Tensor,
               last_idx: torch.Tensor | None = None) -> torch.Tensor:
        """lm_head at each row's LAST REAL position.  last_idx [B] selects
        per-row positions for right-padded batches — without it, shorter
        prompts in a batch would be conditioned on their PAD tail."""
        B, S, d = hidden.shape
        if last_idx is None:
            last = hidden[:, -1, :]
        else:
            last = hidden[torch.arange(B, device=hidden.device),
                          last_idx.to(hidden.device).long()]
        if B <= 64 and self.device != "cpu":
            return ops.lt_linear(last.contiguous(), self.w.lm_head).float()
        return torch.nn.functional.linear(last, self.w.lm_head).float()  # [B, V]

    def prefill(self, tokens: torch.Tensor, cache: KVCache,
                lens: torch.Tensor | None = None) -> torch.Tensor:
        """Prefill the cache; returns last-real-position logits [B, V].
        lens: per-row true prompt lengths for right-padded batches — masks
        pad keys out of attention, starts decode at each row's own length,
        and gathers logits at lens-1 (batch composition no longer changes a
        request's output).  Also picks the decode-attention implementation
        for the upcoming decode steps from the cache fill ratio (prompt
        length is host-side here — no sync)."""
        B, S = tokens.shape
        lens_i = None if lens is None else lens.to(torch.int32)
        hidden = self.forward_hidden(tokens, cache=cache, kv_lens=lens_i)
        if lens_i is None:
            cache.seq_lens[:] = S
            last_idx = None
        else:
            cache.seq_lens.copy_(lens_i)
            last_idx = lens_i.long() - 1
        if self.device != "cpu":
            self.decode_attn_fn = self._decode_attn_impl(
                S / max(cache.max_seq, 1))
        return self.logits(hidden, last_idx)

    def prefill_suffix(self, suffix_tokens: torch.Tensor, cache: KVCache,
                       prefix_len: int,
                       suffix_lens: torch.Tensor | None = None) -> torch.Tensor:
        """Prefill only the suffix against a cache whose rows [0, prefix_len)
        already hold the shared prefix's KV (prefix-KV caching).  Returns
        last-real-position logits; suffix_lens: per-row true suffix lengths
        for right-padded suffix batches."""
        B, S = suffix_tokens.shape
        pos = (prefix_len + torch.arange(S, device=suffix_tokens.device)
               ).unsqueeze(0).expand(B, S)
        lens_abs = (None if suffix_lens is None
                    else (suffix_lens.to(torch.int32) + prefix_len))
        hidden = self.forward_hidden(suffix_tokens, pos=pos, cache=cache,
                                     q_off=prefix_len, kv_lens=lens_abs)
        if lens_abs is None:
            cache.seq_lens[:] = prefix_len + S
            last_idx = None
        else:
            cache.seq_lens.copy_(lens_abs)
            last_idx = suffix_lens.long() - 1
        if self.device != "cpu":
            self.decode_attn_fn = self._decode_attn_impl(
                (prefix_len + S) / max(cache.max_seq, 1))
        else:
            self.decode_attn_fn = None
        return self.logits(hidden, last_idx)

    def decode_step(self, tokens: torch.Tensor, cache: KVCache) -> torch.Tensor:
        """tokens: [B, 1] the latest sampled token; returns logits [B, V]."""
        hidden = self.forward_decode(tokens, cache)
        cache.seq_lens += 1
        return self.logits(hidden)
