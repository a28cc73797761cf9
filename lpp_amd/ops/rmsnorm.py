"""RMSNorm (fwd+bwd).

Kernel manifest row: SURVEY.md §2.7 "RMSNorm x2 + final" — the reference runs
HF ``LlamaRMSNorm`` (imported at models/llama_ds_mp_wrap.py:12, final-norm
LayerSpec at :218).  Memory-bound: target is the HBM roofline (~6.3 TB/s on
MI355X), reached with vectorised bf16x8 loads (guide G13).
"""

from __future__ import annotations

import torch

from . import use_hip, extension


def rmsnorm_ref(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    """Eager reference: full fp32 math, ONE rounding to the input dtype at
    the end (matches the HIP kernel; HF's LlamaRMSNorm double-rounds by
    casting before the weight multiply)."""
    xf = x.float()
    inv = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (xf * inv * weight.float()).to(x.dtype)


class _RMSNormHIP(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps):
        ext = extension()
        y, invrms = ext.rmsnorm_fwd(x, weight, eps)
        ctx.save_for_backward(x, weight, invrms)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, invrms = ctx.saved_tensors
        ext = extension()
        dx, dw = ext.rmsnorm_bwd(dy.contiguous(), x, weight, invrms)
        return dx, dw, None


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6) -> torch.Tensor:
    if use_hip(x):
        return _RMSNormHIP.apply(x.contiguous(), weight, eps)
    return rmsnorm_ref(x, weight, eps)
