"""Reference-op numerics: these functions are the ground truth the HIP
kernels are tested against, so they get their own sanity tests vs plain
torch formulations."""
import math

import pytest
import torch
import torch.nn.functional as F

from comfyui_parallelanything_amd.ops import reference as R


def test_rms_norm_matches_manual():
    x = torch.randn(2, 5, 64)
    w = torch.randn(64)
    out = R.rms_norm(x, w, eps=1e-6)
    ref = x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + 1e-6) * w
    torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-6)


def test_layer_norm_mod_matches_manual():
    x = torch.randn(2, 7, 32)
    scale = torch.randn(2, 32)
    shift = torch.randn(2, 32)
    out = R.layer_norm_mod(x, scale, shift)
    ref = F.layer_norm(x, (32,), eps=1e-6) * (1 + scale[:, None]) + shift[:, None]
    torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-6)


def test_gate_residual_broadcast():
    r = torch.randn(2, 4, 8)
    g = torch.randn(2, 8)
    x = torch.randn(2, 4, 8)
    torch.testing.assert_close(
        R.gate_residual(r, g, x), r + g[:, None] * x
    )


def test_group_norm_silu_matches():
    x = torch.randn(2, 32, 8, 8)
    w = torch.randn(32)
    b = torch.randn(32)
    out = R.group_norm_silu(x, 8, w, b)
    ref = F.silu(F.group_norm(x, 8, w, b, eps=1e-6))
    torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-6)


def test_rope_preserves_norm():
    # rotation: per-pair 2-norm is invariant
    x = torch.randn(1, 2, 16, 32)
    cs = R.rope_freqs(torch.arange(16), 32)
    out = R.rope_apply(x, cs)
    n_in = x.view(1, 2, 16, 16, 2).norm(dim=-1)
    n_out = out.view(1, 2, 16, 16, 2).norm(dim=-1)
    torch.testing.assert_close(n_in, n_out, rtol=1e-5, atol=1e-6)


def test_rope_position_zero_is_identity():
    x = torch.randn(1, 1, 4, 16)
    cs = R.rope_freqs(torch.zeros(4), 16)
    torch.testing.assert_close(R.rope_apply(x, cs), x)


def test_rope_composition():
    # applying pos a then pos b == applying pos a+b
    x = torch.randn(1, 1, 3, 8)
    a = R.rope_freqs(torch.full((3,), 2.0), 8)
    b = R.rope_freqs(torch.full((3,), 5.0), 8)
    ab = R.rope_freqs(torch.full((3,), 7.0), 8)
    out2 = R.rope_apply(R.rope_apply(x, a), b)
    out1 = R.rope_apply(x, ab)
    torch.testing.assert_close(out1, out2, rtol=1e-4, atol=1e-5)


def test_attention_matches_sdpa():
    q = torch.randn(2, 4, 16, 32)
    k = torch.randn(2, 4, 16, 32)
    v = torch.randn(2, 4, 16, 32)
    out = R.attention(q, k, v)
    ref = F.scaled_dot_product_attention(q, k, v)
    torch.testing.assert_close(out, ref, rtol=1e-4, atol=1e-5)


def test_attention_custom_scale():
    q = torch.randn(1, 1, 8, 16)
    k = torch.randn(1, 1, 8, 16)
    v = torch.randn(1, 1, 8, 16)
    out = R.attention(q, k, v, scale=0.5)
    p = torch.softmax(q @ k.transpose(-1, -2) * 0.5, dim=-1)
    torch.testing.assert_close(out, p @ v, rtol=1e-5, atol=1e-6)


def test_timestep_embedding_shape_and_values():
    t = torch.tensor([0.0, 0.5, 1.0])
    emb = R.timestep_embedding(t, 16)
    assert emb.shape == (3, 16)
    # t=0: cos part all ones, sin part all zeros
    torch.testing.assert_close(emb[0, :8], torch.ones(8))
    torch.testing.assert_close(emb[0, 8:], torch.zeros(8))


def test_timestep_embedding_odd_dim_pads():
    emb = R.timestep_embedding(torch.tensor([0.3]), 9)
    assert emb.shape == (1, 9)
    assert emb[0, -1] == 0.0
