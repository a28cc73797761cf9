"""Linear layer whose weight gradient accumulates straight into the
optimizer's flat fp32 buffer (``weight.main_grad``) through a single
hipBLASLt GEMM with beta=1 — no bf16 dW tensor, no separate fp32
accumulation pass (see csrc/wgrad.cpp; SURVEY.md §2.5 grad-accum).

Falls back to stock ``F.linear`` autograd whenever the fused path does not
apply (CPU, eager-forced, no attached optimizer buffer) so all CPU tests
and optimizer-less forward paths are unchanged.
"""

from __future__ import annotations

import torch
import torch.nn.functional as F

from . import extension, force_eager


class _LinearWgradF32(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, weight: torch.nn.Parameter):
        ctx.save_for_backward(x, weight)
        # saved_tensors unwraps the Parameter, so keep the fp32 buffer
        # reference directly (it is optimizer state, not part of the graph)
        ctx.main_grad = weight.main_grad
        return F.linear(x, weight)

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        x, weight = ctx.saved_tensors
        dy = dy.contiguous()
        dx = dy @ weight
        ext = extension()
        ext.wgrad_f32_accum(
            x.reshape(-1, x.shape[-1]).contiguous(),
            dy.reshape(-1, dy.shape[-1]),
            ctx.main_grad,
        )
        return dx, None


def lp_linear(x: torch.Tensor, weight: torch.nn.Parameter) -> torch.Tensor:
    """F.linear with fused-fp32-wgrad backward when the optimizer has
    attached ``main_grad`` (training on GPU); plain autograd otherwise."""
    if (
        x.is_cuda
        and x.dtype is torch.bfloat16
        and torch.is_grad_enabled()
        and weight.requires_grad
        and hasattr(weight, "main_grad")
        and not force_eager()
    ):
        return _LinearWgradF32.apply(x, weight)
    return F.linear(x, weight)


class LPLinear(torch.nn.Linear):
    """Drop-in nn.Linear (bias-free) using the fused-wgrad path."""

    def __init__(self, in_features: int, out_features: int, bias: bool = False):
        assert not bias, "LLaMA projections are bias-free"
        super().__init__(in_features, out_features, bias=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return lp_linear(x, self.weight)
