"""Eager reference-op unit tests (closed-form / manual-formula oracles).

These same references are the numerics oracle for the HIP kernels
(tests/test_gpu_kernels.py compares kernel vs these at fp32 tolerance).
"""

import math

import pytest
import torch

from lpp_amd.ops.attention import causal_attention_ref
from lpp_amd.ops.cross_entropy import shifted_cross_entropy_ref
from lpp_amd.ops.rmsnorm import rmsnorm_ref
from lpp_amd.ops.rope import apply_rope_ref, build_rope_cache
from lpp_amd.ops.swiglu import swiglu_ref

torch.manual_seed(0)


def test_rmsnorm_manual():
    x = torch.randn(2, 5, 16)
    w = torch.randn(16)
    y = rmsnorm_ref(x, w, eps=1e-6)
    for b in range(2):
        for s in range(5):
            row = x[b, s]
            rms = math.sqrt((row.double() ** 2).mean().item() + 1e-6)
            exp = row / rms * w
            assert torch.allclose(y[b, s], exp, atol=1e-5)


def test_rope_rotation_preserves_norm_and_composes():
    cos, sin = build_rope_cache(seq_len=32, head_dim=8, theta=10000.0, device="cpu")
    x = torch.randn(1, 32, 2, 8)
    y = apply_rope_ref(x, cos, sin)
    # rotation preserves pairwise norms
    x2 = x[..., :4] ** 2 + x[..., 4:] ** 2
    y2 = y[..., :4] ** 2 + y[..., 4:] ** 2
    assert torch.allclose(x2, y2, atol=1e-4)
    # position 0 is identity
    assert torch.allclose(y[:, 0], x[:, 0], atol=1e-6)


def test_rope_relative_property():
    """q.k after RoPE depends only on relative distance."""
    D = 16
    cos, sin = build_rope_cache(64, D, 10000.0, "cpu")
    q = torch.randn(1, 1, 1, D)
    k = torch.randn(1, 1, 1, D)
    def dot_at(pq, pk):
        qr = apply_rope_ref(q, cos, sin, pos_offset=pq)
        kr = apply_rope_ref(k, cos, sin, pos_offset=pk)
        return (qr * kr).sum().item()
    assert abs(dot_at(5, 3) - dot_at(12, 10)) < 1e-4


def test_rope_matches_hf_convention():
    transformers = pytest.importorskip("transformers")
    from transformers.models.llama.modeling_llama import apply_rotary_pos_emb

    B, S, H, D = 2, 16, 4, 32
    x = torch.randn(B, S, H, D)
    cos, sin = build_rope_cache(S, D, 10000.0, "cpu")
    ours = apply_rope_ref(x, cos, sin)
    # HF wants [B, H, S, D] and full-width cos/sin [B, S, D]
    cos_full = torch.cat([cos, cos], dim=-1)[None].expand(B, S, D)
    sin_full = torch.cat([sin, sin], dim=-1)[None].expand(B, S, D)
    hf_q, _ = apply_rotary_pos_emb(
        x.transpose(1, 2), x.transpose(1, 2), cos_full, sin_full
    )
    assert torch.allclose(ours, hf_q.transpose(1, 2), atol=1e-5)


def test_swiglu_manual():
    g = torch.randn(64)
    u = torch.randn(64)
    y = swiglu_ref(g, u)
    exp = g * torch.sigmoid(g) * u
    assert torch.allclose(y, exp, atol=1e-6)


def test_shifted_ce_matches_manual():
    B, S, V = 2, 8, 31
    logits = torch.randn(B, S, V)
    labels = torch.randint(0, V, (B, S))
    labels[0, 3] = -100
    loss = shifted_cross_entropy_ref(logits, labels)
    man = torch.nn.functional.cross_entropy(
        logits[:, :-1].reshape(-1, V), labels[:, 1:].reshape(-1), ignore_index=-100
    )
    assert torch.allclose(loss, man, atol=1e-6)


def test_shifted_ce_ignores_prompt():
    B, S, V = 1, 6, 11
    logits = torch.randn(B, S, V)
    labels = torch.full((B, S), -100)
    labels[0, -1] = 3
    loss = shifted_cross_entropy_ref(logits, labels)
    man = torch.nn.functional.cross_entropy(logits[0, -2][None], torch.tensor([3]))
    assert torch.allclose(loss, man, atol=1e-6)


def test_causal_attention_matches_masked_eager():
    B, S, H, D = 2, 12, 4, 16
    q = torch.randn(B, S, H, D)
    k = torch.randn(B, S, H, D)
    v = torch.randn(B, S, H, D)
    out = causal_attention_ref(q, k, v)
    # manual eager with explicit mask
    qt, kt, vt = (t.transpose(1, 2) for t in (q, k, v))
    scores = qt @ kt.transpose(-1, -2) / math.sqrt(D)
    mask = torch.triu(torch.ones(S, S, dtype=torch.bool), diagonal=1)
    scores = scores.masked_fill(mask, float("-inf"))
    exp = (scores.softmax(-1) @ vt).transpose(1, 2)
    assert torch.allclose(out, exp, atol=1e-5)


def test_causal_attention_gqa():
    B, S, H, Hkv, D = 1, 8, 4, 2, 8
    q = torch.randn(B, S, H, D)
    k = torch.randn(B, S, Hkv, D)
    v = torch.randn(B, S, Hkv, D)
    out = causal_attention_ref(q, k, v)
    assert out.shape == (B, S, H, D)
    # head 0,1 use kv head 0
    k_exp = k.repeat_interleave(2, dim=2)
    v_exp = v.repeat_interleave(2, dim=2)
    exp = causal_attention_ref(q, k_exp, v_exp)
    assert torch.allclose(out, exp, atol=1e-6)
