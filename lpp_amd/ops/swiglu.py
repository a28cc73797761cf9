"""Fused SwiGLU: y = silu(gate) * up  (fwd+bwd).

Kernel manifest row: SURVEY.md §2.7 "MLP gate/up + SiLU(.)" — the reference
runs HF LlamaMLP's unfused silu-then-mul.  Fusing the elementwise pair halves
HBM traffic on the [B*S, I] intermediate (I=22016 for 65B), the largest
activation tensor in the layer.
"""

from __future__ import annotations

import torch

from . import use_hip, extension


def swiglu_ref(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    gf = gate.float()
    return (gf * torch.sigmoid(gf) * up.float()).to(gate.dtype)


class _SwiGLUHIP(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gate, up):
        ext = extension()
        y = ext.swiglu_fwd(gate, up)
        ctx.save_for_backward(gate, up)
        return y

    @staticmethod
    def backward(ctx, dy):
        gate, up = ctx.saved_tensors
        ext = extension()
        dgate, dup = ext.swiglu_bwd(dy.contiguous(), gate, up)
        return dgate, dup


def swiglu(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    if use_hip(gate):
        return _SwiGLUHIP.apply(gate.contiguous(), up.contiguous())
    return swiglu_ref(gate, up)
