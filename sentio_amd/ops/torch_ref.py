"""Plain-PyTorch fp32 reference implementations of every HIP op.

These are (a) the CPU execution path for tests and the hermetic config, and
(b) the numerics references the GPU tests compare the gfx950 kernels against
(driver contract: numerics tests compare HIP kernels vs plain PyTorch fp32).
Shapes/semantics documented per-op; the HIP kernels implement exactly these.
"""

from __future__ import annotations

import math

import torch


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    """y = x / rms(x) * weight over the last dim."""
    xf = x.float()
    rms = xf.pow(2).mean(dim=-1, keepdim=True).add(eps).rsqrt()
    return (xf * rms * weight.float()).to(x.dtype)


def rmsnorm_residual(
    x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5
) -> tuple[torch.Tensor, torch.Tensor]:
    """h = x + residual;  y = rmsnorm(h).  Returns (y, h) — the fused
    residual-add + norm used between transformer blocks."""
    h = (x.float() + residual.float())
    rms = h.pow(2).mean(dim=-1, keepdim=True).add(eps).rsqrt()
    y = (h * rms * weight.float()).to(x.dtype)
    return y, h.to(x.dtype)


def rope_tables(
    max_seq: int, head_dim: int, base: float = 500000.0, device: str = "cpu"
) -> tuple[torch.Tensor, torch.Tensor]:
    """Precomputed cos/sin tables [max_seq, head_dim/2] (fp32).
    base=500000 is the Llama-3 rope theta."""
    inv = 1.0 / (
        base ** (torch.arange(0, head_dim, 2, device=device).float() / head_dim)
    )
    t = torch.arange(max_seq, device=device).float()
    freqs = torch.outer(t, inv)
    return freqs.cos(), freqs.sin()


def rope_apply(
    x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor, pos: torch.Tensor
) -> torch.Tensor:
    """Rotate pairs (x[2i], x[2i+1]) by position angles.

    x: [B, S, H, D]; cos/sin: [max_seq, D/2]; pos: [B, S] int32 positions.
    """
    B, S, H, D = x.shape
    c = cos[pos.long()].unsqueeze(2)  # [B,S,1,D/2]
    s = sin[pos.long()].unsqueeze(2)
    xf = x.float().view(B, S, H, D // 2, 2)
    x0, x1 = xf[..., 0], xf[..., 1]
    out = torch.stack((x0 * c - x1 * s, x0 * s + x1 * c), dim=-1)
    return out.view(B, S, H, D).to(x.dtype)


def swiglu(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    """silu(gate) * up."""
    return (torch.nn.functional.silu(gate.float()) * up.float()).to(gate.dtype)


def softmax(x: torch.Tensor, dim: int = -1) -> torch.Tensor:
    return torch.softmax(x.float(), dim=dim).to(x.dtype)


def attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    causal: bool = True,
    scale: float | None = None,
    kv_lens: torch.Tensor | None = None,
) -> torch.Tensor:
    """Reference multi-head attention with GQA.

    q: [B, S, H, D];  k, v: [B, S, Hkv, D] with H % Hkv == 0.
    kv_lens: optional [B] int — right-padding mask: keys at position >= len
    are ignored (encoder batches).  Returns [B, S, H, D].  Computed in fp32.
    """
    B, S, H, D = q.shape
    Hkv = k.shape[2]
    rep = H // Hkv
    scale = scale if scale is not None else 1.0 / math.sqrt(D)
    qf = q.float().permute(0, 2, 1, 3)  # [B,H,S,D]
    kf = k.float().permute(0, 2, 1, 3).repeat_interleave(rep, dim=1)
    vf = v.float().permute(0, 2, 1, 3).repeat_interleave(rep, dim=1)
    scores = (qf @ kf.transpose(-1, -2)) * scale  # [B,H,S,S]
    if causal:
        mask = torch.full((S, S), float("-inf"), device=q.device).triu(1)
        scores = scores + mask
    if kv_lens is not None:
        key_valid = (
            torch.arange(S, device=q.device).view(1, 1, 1, S)
            < kv_lens.view(B, 1, 1, 1)
        )
        scores = scores.masked_fill(~key_valid, float("-inf"))
    probs = torch.softmax(scores, dim=-1)
    out = probs @ vf  # [B,H,S,D]
    return out.permute(0, 2, 1, 3).contiguous().to(q.dtype)


def decode_attention(
    q: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    seq_lens: torch.Tensor,
    scale: float | None = None,
) -> torch.Tensor:
    """Single-token decode attention over a contiguous KV cache.

    q: [B, H, D] (the new token's query);
    k_cache/v_cache: [B, Hkv, Smax, D]; seq_lens: [B] valid lengths.
    Returns [B, H, D].
    """
    B, H, D = q.shape
    Hkv = k_cache.shape[1]
    rep = H // Hkv
    scale = scale if scale is not None else 1.0 / math.sqrt(D)
    out = torch.empty(B, H, D, dtype=q.dtype, device=q.device)
    for b in range(B):
        s = int(seq_lens[b])
        kf = k_cache[b, :, :s].float().repeat_interleave(rep, dim=0)  # [H,s,D]
        vf = v_cache[b, :, :s].float().repeat_interleave(rep, dim=0)
        qs = q[b].float().unsqueeze(1)  # [H,1,D]
        p = torch.softmax((qs @ kf.transpose(-1, -2)) * scale, dim=-1)
        out[b] = (p @ vf).squeeze(1).to(q.dtype)
    return out


def mean_pool_l2norm(hidden: torch.Tensor, mask: torch.Tensor) -> torch.Tensor:
    """Masked mean-pool over sequence then L2-normalize.

    hidden: [B, S, D]; mask: [B, S] (1 = real token). Returns [B, D] fp32.
    """
    m = mask.float().unsqueeze(-1)
    summed = (hidden.float() * m).sum(dim=1)
    counts = m.sum(dim=1).clamp_min(1.0)
    pooled = summed / counts
    return pooled / pooled.norm(dim=-1, keepdim=True).clamp_min(1e-12)


def cosine_topk(
    q: torch.Tensor, mat: torch.Tensor, k: int
) -> tuple[torch.Tensor, torch.Tensor]:
    """q: [B, D] (normalized), mat: [N, D] (normalized rows).
    Returns (values [B,k] fp32, indices [B,k] int64)."""
    scores = q.float() @ mat.float().T
    return torch.topk(scores, k, dim=1)


def bm25_score(
    term_ids: torch.Tensor,
    indptr: torch.Tensor,
    post_doc: torch.Tensor,
    post_tf: torch.Tensor,
    idf: torch.Tensor,
    doc_len: torch.Tensor,
    n_docs: int,
    k1: float,
    b: float,
    avgdl: float,
    plus_delta: float = 0.0,
) -> torch.Tensor:
    """Dense BM25 scores over all docs for one query's term ids."""
    scores = torch.zeros(n_docs, dtype=torch.float32, device=term_ids.device)
    norm_den = k1 * (1.0 - b + b * doc_len.float() / avgdl)
    for tid in term_ids.tolist():
        lo, hi = int(indptr[tid]), int(indptr[tid + 1])
        docs = post_doc[lo:hi].long()
        tf = post_tf[lo:hi].float()
        contrib = tf * (k1 + 1.0) / (tf + norm_den[docs])
        if plus_delta:
            contrib = contrib + plus_delta
        scores.index_add_(0, docs, float(idf[tid]) * contrib)
    return scores


def sample_token(
    logits: torch.Tensor, temperature: float, generator: torch.Generator | None = None
) -> torch.Tensor:
    """logits: [B, V] → [B] int64 sampled (or argmax at T==0)."""
    if temperature <= 0.0:
        return logits.float().argmax(dim=-1)
    probs = torch.softmax(logits.float() / temperature, dim=-1)
    return torch.multinomial(probs, 1, generator=generator).squeeze(-1)


def decode_qkv_prep(
    qkv: torch.Tensor, k_cache: torch.Tensor, v_cache: torch.Tensor,
    cos: torch.Tensor, sin: torch.Tensor, seq_lens: torch.Tensor
) -> torch.Tensor:
    """CPU reference of the fused decode head prep (see ops.decode_qkv_prep):
    splits the raw S=1 QKV projection, RoPE-rotates q and k, writes k/v into
    the caches at row seq_lens[b], returns q [B, H, D]."""
    B = qkv.shape[0]
    Hkv, smax, D = k_cache.shape[1], k_cache.shape[2], k_cache.shape[3]
    H = qkv.shape[1] // D - 2 * Hkv
    pos = seq_lens.long().view(B, 1)
    q = qkv[:, : H * D].view(B, 1, H, D)
    k = qkv[:, H * D : (H + Hkv) * D].view(B, 1, Hkv, D)
    v = qkv[:, (H + Hkv) * D :].view(B, 1, Hkv, D)
    q = rope_apply(q, cos, sin, pos)
    k = rope_apply(k, cos, sin, pos)
    b_idx = torch.arange(B)
    k_cache[b_idx, :, seq_lens.long()] = k.view(B, Hkv, D).to(k_cache.dtype)
    v_cache[b_idx, :, seq_lens.long()] = v.view(B, Hkv, D).to(v_cache.dtype)
    return q.view(B, H, D)


def swiglu_packed(gu: torch.Tensor) -> torch.Tensor:
    """gu [..., 2F] rows packed [gate | up] → silu(gate) * up, [..., F]."""
    f = gu.shape[-1] // 2
    return swiglu(gu[..., :f], gu[..., f:])


def attention_cache(
    q: torch.Tensor, k_cache: torch.Tensor, v_cache: torch.Tensor,
    kv_lens: torch.Tensor, q_off: int, scale: float | None = None,
) -> torch.Tensor:
    """Suffix-against-cache causal attention (prefix-KV-cached prefill).
    q: [B, S_suf, H, D] at absolute positions q_off + i;
    k_cache/v_cache: [B, Hkv, Smax, D]; kv_lens: [B] absolute lengths."""
    B, S, H, D = q.shape
    Hkv = k_cache.shape[1]
    rep = H // Hkv
    scale = scale if scale is not None else 1.0 / math.sqrt(D)
    out = torch.empty_like(q, dtype=torch.float32)
    for b in range(B):
        L = int(kv_lens[b])
        kf = k_cache[b, :, :L].float().repeat_interleave(rep, dim=0)  # [H,L,D]
        vf = v_cache[b, :, :L].float().repeat_interleave(rep, dim=0)
        qf = q[b].float().permute(1, 0, 2)                            # [H,S,D]
        sc = (qf @ kf.transpose(-1, -2)) * scale                      # [H,S,L]
        pos = q_off + torch.arange(S).view(1, S, 1)
        key = torch.arange(L).view(1, 1, L)
        sc = sc.masked_fill(key > pos, float("-inf"))
        p = torch.softmax(sc, dim=-1)
        out[b] = (p @ vf).permute(1, 0, 2)
    return out.to(q.dtype)
