"""CPU reference ops: numerics sanity vs hand-computed math.
(The GPU kernels are compared against these same references in
tests/test_ops_gpu.py, per the driver's numerics-test contract.)"""

import math

import pytest
import torch

from sentio_amd.ops import torch_ref as R


def test_rmsnorm_matches_manual():
    x = torch.randn(4, 8, dtype=torch.float32)
    w = torch.randn(8)
    y = R.rmsnorm(x, w, eps=1e-5)
    for i in range(4):
        rms = math.sqrt(float((x[i] ** 2).mean()) + 1e-5)
        manual = x[i] / rms * w
        torch.testing.assert_close(y[i], manual, rtol=1e-5, atol=1e-5)


def test_rmsnorm_residual_returns_sum_and_norm():
    x = torch.randn(2, 3, 8)
    res = torch.randn(2, 3, 8)
    w = torch.ones(8)
    y, h = R.rmsnorm_residual(x, res, w)
    torch.testing.assert_close(h, x + res)
    torch.testing.assert_close(y, R.rmsnorm(x + res, w))


def test_rope_rotation_preserves_norm_and_is_positional():
    cos, sin = R.rope_tables(32, 8, base=10000.0)
    x = torch.randn(1, 4, 2, 8)
    pos = torch.arange(4).unsqueeze(0)
    y = R.rope_apply(x, cos, sin, pos)
    # rotation preserves pair norms
    torch.testing.assert_close(
        y.view(1, 4, 2, 4, 2).norm(dim=-1), x.view(1, 4, 2, 4, 2).norm(dim=-1),
        rtol=1e-5, atol=1e-5)
    # position 0 is identity
    torch.testing.assert_close(y[:, 0], x[:, 0], rtol=1e-6, atol=1e-6)
    # nonzero positions rotate
    assert not torch.allclose(y[:, 1], x[:, 1])


def test_swiglu():
    g = torch.randn(5, 7)
    u = torch.randn(5, 7)
    y = R.swiglu(g, u)
    torch.testing.assert_close(y, torch.nn.functional.silu(g) * u)


def test_attention_causal_matches_manual_single_head():
    B, S, D = 1, 5, 4
    q = torch.randn(B, S, 1, D)
    k = torch.randn(B, S, 1, D)
    v = torch.randn(B, S, 1, D)
    out = R.attention(q, k, v, causal=True)
    for t in range(S):
        scores = (q[0, t, 0] @ k[0, : t + 1, 0].T) / math.sqrt(D)
        probs = torch.softmax(scores, dim=-1)
        manual = probs @ v[0, : t + 1, 0]
        torch.testing.assert_close(out[0, t, 0], manual, rtol=1e-5, atol=1e-5)


def test_attention_gqa_equals_repeated_kv():
    B, S, H, Hkv, D = 2, 6, 4, 2, 8
    q = torch.randn(B, S, H, D)
    k = torch.randn(B, S, Hkv, D)
    v = torch.randn(B, S, Hkv, D)
    out_gqa = R.attention(q, k, v, causal=True)
    k_rep = k.repeat_interleave(H // Hkv, dim=2)
    v_rep = v.repeat_interleave(H // Hkv, dim=2)
    out_full = R.attention(q, k_rep, v_rep, causal=True)
    torch.testing.assert_close(out_gqa, out_full)


def test_decode_attention_matches_prefill_last_position():
    B, S, H, Hkv, D = 2, 7, 4, 2, 8
    q_all = torch.randn(B, S, H, D)
    k_all = torch.randn(B, S, Hkv, D)
    v_all = torch.randn(B, S, Hkv, D)
    full = R.attention(q_all, k_all, v_all, causal=True)
    k_cache = k_all.permute(0, 2, 1, 3).contiguous()
    v_cache = v_all.permute(0, 2, 1, 3).contiguous()
    dec = R.decode_attention(q_all[:, -1].permute(0, 1, 2), k_cache, v_cache,
                             torch.full((B,), S, dtype=torch.int32))
    torch.testing.assert_close(dec, full[:, -1], rtol=1e-5, atol=1e-5)


def test_mean_pool_l2norm_masks_padding():
    h = torch.randn(2, 4, 8)
    mask = torch.tensor([[1, 1, 0, 0], [1, 1, 1, 1]])
    out = R.mean_pool_l2norm(h, mask)
    manual0 = h[0, :2].mean(dim=0)
    manual0 = manual0 / manual0.norm()
    torch.testing.assert_close(out[0], manual0, rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(out.norm(dim=-1), torch.ones(2), rtol=1e-5, atol=1e-5)


def test_cosine_topk():
    q = torch.nn.functional.normalize(torch.randn(3, 16), dim=1)
    m = torch.nn.functional.normalize(torch.randn(50, 16), dim=1)
    vals, idx = R.cosine_topk(q, m, 5)
    full = q @ m.T
    want_vals, want_idx = torch.topk(full, 5, dim=1)
    torch.testing.assert_close(vals, want_vals)
    assert torch.equal(idx, want_idx)


def test_sample_token_greedy_and_temperature():
    logits = torch.tensor([[0.0, 10.0, 0.0], [5.0, 0.0, 0.0]])
    g = R.sample_token(logits, 0.0)
    assert g.tolist() == [1, 0]
    gen = torch.Generator().manual_seed(0)
    s = R.sample_token(logits, 1.0, gen)
    assert s.shape == (2,)
