"""Linear layer whose weight gradient accumulates straight into the
optimizer's flat fp32 buffer (``weight.main_grad``) through a single
hipBLASLt GEMM with beta=1 — no bf16 dW tensor, no separate fp32
accumulation pass (see csrc/wgrad.cpp; SURVEY.md §2.5 grad-accum).

Falls back to stock ``F.linear`` autograd whenever the fused path does not
apply (CPU, eager-forced, no attached optimizer buffer) so all CPU tests
and optimizer-less forward paths are unchanged.
"""

from __future__ import annotations

import os

import torch
import torch.nn.functional as F

from . import extension, force_eager


class _XTCache:
    """One-entry memo for the backward-side activation transpose.

    q/k/v (and gate/up) share the same input activation and their backwards
    run consecutively, so the xT produced for the first projection is
    reused for its siblings.  Keyed by tensor identity with a STRONG
    reference held, so the storage cannot be recycled under the key."""

    __slots__ = ("x", "xT")

    def __init__(self):
        self.x = None
        self.xT = None

    def get(self, ext, x2: torch.Tensor) -> torch.Tensor:
        # The held strong reference keeps the storage alive, so a matching
        # (data_ptr, shape, dtype) can only be the same activation.
        if (self.x is not None and self.x.data_ptr() == x2.data_ptr()
                and self.x.shape == x2.shape and self.x.dtype == x2.dtype):
            return self.xT
        xT = ext.transpose2d(x2)
        self.x = x2
        self.xT = xT
        return xT

    def clear(self):
        self.x = None
        self.xT = None


_xt_cache = _XTCache()


def _wgrad_pre_enabled() -> bool:
    return os.environ.get("LPP_WGRAD_PRE", "1") == "1"


def _wgrad_bf16d_enabled() -> bool:
    """bf16-D wgrad GEMM + separate fp32 accumulate (default): hipBLASLt's
    bf16-D solution pool is 15-20% faster than fp32-D at the 65B shapes and
    the accumulate pass is ~5% of the GEMM.  Numerics match the reference
    stack's own flow (bf16 dW per microbatch, fp32 accumulation across
    microbatches — SURVEY.md §2.5).  Set LPP_WGRAD_BF16D=0 for the exact
    fp32-D GEMM epilogue."""
    return os.environ.get("LPP_WGRAD_BF16D", "1") == "1"


def _dgrad_wt_enabled() -> bool:
    return os.environ.get("LPP_DGRAD_WT", "1") == "1"


def _weight_t(ext, weight: torch.nn.Parameter) -> torch.Tensor:
    """Cached transposed weight for the dgrad GEMM.

    dx = dy @ W is the k-strided hipBLASLt class (the contraction dim runs
    down W's rows) and measures 8-21% below the k-contiguous class at the
    65B shapes (profiles/r02_gemm_probe2.txt).  W only changes at the
    optimizer step, so one transposed copy per step turns every dgrad into
    the fast class:  dx = dy @ (W^T)^T with W^T materialised [in, out].
    The engine invalidates the cache after each optimizer step
    (engine._optimizer_step -> invalidate_weight_transposes)."""
    wt = getattr(weight, "_wt", None)
    if wt is None:
        wt = ext.transpose2d(weight.data)
        weight._wt = wt
    return wt


def invalidate_weight_transposes(module: torch.nn.Module) -> None:
    """Drop cached W^T copies (call after any in-place weight update)."""
    for p in module.parameters():
        if hasattr(p, "_wt"):
            p._wt = None


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
        ext = extension()
        if _dgrad_wt_enabled():
            dx = dy @ _weight_t(ext, weight).t()
        else:
            dx = dy @ weight
        x2 = x.reshape(-1, x.shape[-1]).contiguous()
        dy2 = dy.reshape(-1, dy.shape[-1])
        if _wgrad_pre_enabled():
            # Pre-transposed formulation: both GEMM operands k-contiguous
            # (the fast hipBLASLt class, +25-40% over the natural TN
            # layout) at the cost of two HBM-speed LDS-tiled transposes;
            # the x transpose is shared across sibling projections.
            xT = _xt_cache.get(ext, x2)
            dyT = ext.transpose2d(dy2)
            if _wgrad_bf16d_enabled():
                ext.accum_bf16_f32(ctx.main_grad.view(-1),
                                   torch.matmul(dyT, xT.t()).view(-1))
            else:
                ext.wgrad_f32_accum_pre(xT, dyT, ctx.main_grad)
        else:
            ext.wgrad_f32_accum(x2, dy2, ctx.main_grad)
        # This path bypasses autograd's grad accumulation, so the engine's
        # DP-bucket overlap (post-accumulate hooks) must be notified here.
        cb = getattr(weight, "_on_accumulate", None)
        if cb is not None:
            cb(weight)
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
