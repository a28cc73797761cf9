"""Causal self-attention core.

The reference runs *materialised-mask eager attention* — the [B,1,S,S] fp16
additive mask is built host-side in the collator (data/flan.py:194-243) and
shipped through every pipeline stage; flash attention was documented broken
under DeepSpeed-PP (README.md:141-143).  This module fixes that design:
the mask is implicit (causal) — nothing S^2-shaped is ever materialised,
shipped over xGMI, or saved for backward.

Dispatch:
- HIP flash-style kernel (gfx950 MFMA, online softmax) when available.
- Otherwise torch SDPA with ``is_causal=True`` (CPU tests / A-B baseline).

Layout contract: q [B, S, H, D], k/v [B, S, Hkv, D]; returns [B, S, H, D].
"""

from __future__ import annotations

import torch

from . import use_hip, extension, force_eager


def causal_attention_ref(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor) -> torch.Tensor:
    B, S, H, D = q.shape
    Hkv = k.shape[2]
    qt = q.transpose(1, 2)  # [B,H,S,D]
    kt = k.transpose(1, 2)
    vt = v.transpose(1, 2)
    if Hkv != H:
        rep = H // Hkv
        kt = kt.repeat_interleave(rep, dim=1)
        vt = vt.repeat_interleave(rep, dim=1)
    out = torch.nn.functional.scaled_dot_product_attention(qt, kt, vt, is_causal=True)
    return out.transpose(1, 2).contiguous()


def _hip_supported(q: torch.Tensor) -> bool:
    """HIP flash kernels cover the production shape: bf16, head_dim 128."""
    if force_eager():
        return False
    if q.dtype is not torch.bfloat16 or q.shape[-1] != 128:
        return False
    ext = extension()  # raises loudly if the native path is missing on GPU
    return hasattr(ext, "attention_fwd") and hasattr(ext, "attention_bwd")


class _FlashAttnHIP(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v):
        ext = extension()
        o, lse = ext.attention_fwd(q, k, v)
        ctx.save_for_backward(q, k, v, o, lse)
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse = ctx.saved_tensors
        ext = extension()
        dq, dk, dv = ext.attention_bwd(do.contiguous(), q, k, v, o, lse)
        return dq, dk, dv


def causal_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor) -> torch.Tensor:
    if use_hip(q) and _hip_supported(q):
        return _FlashAttnHIP.apply(q.contiguous(), k.contiguous(), v.contiguous())
    return causal_attention_ref(q, k, v)
