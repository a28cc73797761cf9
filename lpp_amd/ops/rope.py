"""Rotary position embedding (RoPE), fwd+bwd.

Kernel manifest row: SURVEY.md §2.7 "RoPE apply (q,k)".  The reference
threads ``position_ids`` through every stage tuple
(models/llama_ds_mp_wrap.py:25,37,148) and lets HF apply RoPE; our engine
regenerates positions per stage (never shipped over xGMI) and applies RoPE
with a host-precomputed cos/sin table (guide Appendix B: on-device trig
turns a memory-bound op VALU-bound — precompute on host, load as fp32).

Layout: q/k are [B, S, H, D] (head-last, D contiguous) — pairs (d, d+D/2)
rotated, matching HF's rotate_half convention so converted HF checkpoints
produce identical logits.
"""

from __future__ import annotations

from typing import Tuple

import torch

from . import use_hip, extension

_CACHE: dict = {}


def build_rope_cache(
    seq_len: int, head_dim: int, theta: float, device, dtype=torch.float32
) -> Tuple[torch.Tensor, torch.Tensor]:
    """cos/sin tables [S, D/2], fp32, cached per (S, D, theta, device)."""
    key = (seq_len, head_dim, theta, str(device))
    hit = _CACHE.get(key)
    if hit is not None:
        return hit
    inv_freq = 1.0 / (
        theta ** (torch.arange(0, head_dim, 2, device=device, dtype=torch.float32) / head_dim)
    )
    t = torch.arange(seq_len, device=device, dtype=torch.float32)
    freqs = torch.outer(t, inv_freq)  # [S, D/2]
    out = (freqs.cos().contiguous(), freqs.sin().contiguous())
    _CACHE[key] = out
    return out


def apply_rope_ref(
    x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor, pos_offset: int = 0
) -> torch.Tensor:
    """x: [B, S, H, D]. rotate_half convention (HF): pairs (d, d + D/2)."""
    B, S, H, D = x.shape
    c = cos[pos_offset : pos_offset + S].view(1, S, 1, D // 2).to(torch.float32)
    s = sin[pos_offset : pos_offset + S].view(1, S, 1, D // 2).to(torch.float32)
    xf = x.float()
    x1, x2 = xf[..., : D // 2], xf[..., D // 2 :]
    out = torch.cat((x1 * c - x2 * s, x2 * c + x1 * s), dim=-1)
    return out.to(x.dtype)


class _RopeHIP(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, cos, sin, pos_offset):
        ext = extension()
        y = ext.rope_fwd(x, cos, sin, pos_offset)
        ctx.save_for_backward(cos, sin)
        ctx.pos_offset = pos_offset
        return y

    @staticmethod
    def backward(ctx, dy):
        cos, sin = ctx.saved_tensors
        ext = extension()
        # RoPE is a rotation; backward rotates by -theta == sin sign flip.
        dx = ext.rope_bwd(dy.contiguous(), cos, sin, ctx.pos_offset)
        return dx, None, None, None


def apply_rope(
    x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor, pos_offset: int = 0
) -> torch.Tensor:
    if use_hip(x):
        return _RopeHIP.apply(x.contiguous(), cos, sin, pos_offset)
    return apply_rope_ref(x, cos, sin, pos_offset)


@torch.no_grad()
def apply_rope_cs(x: torch.Tensor, c: torch.Tensor, s: torch.Tensor) -> torch.Tensor:
    """Apply RoPE with pre-gathered per-row cos/sin ([N,1,1,D/2]) — the
    gather is identical for every layer of a decode tick, so the serving
    engine hoists it (see apply_rope_positions)."""
    D = x.shape[-1]
    xf = x.float()
    x1, x2 = xf[..., : D // 2], xf[..., D // 2 :]
    return torch.cat((x1 * c - x2 * s, x2 * c + x1 * s), dim=-1).to(x.dtype)


@torch.no_grad()
def apply_rope_positions(
    x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor, positions: torch.Tensor
) -> torch.Tensor:
    """Ragged variant for continuous-batching decode: x [N, 1, H, D] with a
    DIFFERENT position per row (positions [N] long).  Eager on purpose —
    single-token decode is tiny and latency-bound."""
    N, S, H, D = x.shape
    assert S == 1, "positions variant is a single-token decode contract"
    c = cos[positions].view(N, 1, 1, D // 2).to(torch.float32)
    s = sin[positions].view(N, 1, 1, D // 2).to(torch.float32)
    return apply_rope_cs(x, c, s)
