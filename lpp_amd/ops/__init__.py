"""Hot-path ops with two implementations:

- **HIP/CDNA4 kernels** (``lpp_amd/ops/csrc``) — the path that runs on
  MI355X.  Hand-written for gfx950: 64-wide wavefronts, vectorised bf16
  (short4/short8) loads, LDS where reuse exists, MFMA for matmul-shaped work.
- **eager PyTorch reference** — numerically-plain fp32-upcast composition of
  stock ops.  This is the CPU path, the numerics oracle for every kernel
  test, and the A/B baseline (set ``LPP_FORCE_EAGER=1``).

On a GPU box the HIP extension is REQUIRED: if a CUDA tensor reaches an op
and the extension failed to load, we raise instead of silently falling back
(the whole point of the framework is the native path).

The per-op kernel manifest mirrors SURVEY.md §2.7 (the reference's compute
graph is HF LlamaDecoderLayer + loss_fn; models/llama_ds_mp_wrap.py:8-13,
105-116 — it ships no kernels of its own): rmsnorm, rope, swiglu, fused
shifted cross-entropy, fused AdamW, flash causal attention fwd+bwd, and
the hipBLASLt fp32-accumulating weight-gradient GEMM (linear.py).
"""

from __future__ import annotations

import os

import torch

_EXT = None
_EXT_ERR: str | None = "not loaded yet"


def _load_extension():
    global _EXT, _EXT_ERR
    if _EXT is not None:
        return _EXT
    try:
        from . import build as _build

        _EXT = _build.load_extension()
        _EXT_ERR = None
        if torch.cuda.is_available():
            _pin_wgrad_algos(_EXT)
    except Exception as e:  # pragma: no cover - exercised on GPU box
        _EXT = None
        _EXT_ERR = f"{type(e).__name__}: {e}"
    return _EXT


def _pin_wgrad_algos(ext) -> None:
    """Pin committed hipBLASLt solution indices for the wgrad GEMM shapes
    (lpp_amd/ops/wgrad_algos.json, produced by scripts/wgrad_tune.py's
    exhaustive device-timed sweep).  Indices are library-version-specific;
    the JSON records the hipBLASLt version it was tuned on and is skipped
    on mismatch (falling back to first-use heuristic timing)."""
    import json

    path = os.path.join(os.path.dirname(__file__), "wgrad_algos.json")
    if not os.path.exists(path):
        return
    try:
        data = json.load(open(path))
        for entry in data.get("shapes", []):
            ext.wgrad_set_algo(entry["T"], entry["in"], entry["out"], entry["index"],
                               entry.get("kind", 0))
    except Exception as e:  # pragma: no cover
        import logging

        logging.getLogger(__name__).warning(
            "wgrad algo pinning skipped: %s", e)


def force_eager() -> bool:
    return os.environ.get("LPP_FORCE_EAGER", "0") == "1"


def extension():
    """The loaded HIP extension module, or raise with the load error."""
    ext = _load_extension()
    if ext is None:
        raise RuntimeError(
            "lpp_amd HIP extension is required on GPU but failed to load: "
            f"{_EXT_ERR}. Build it with `python -m lpp_amd.ops.build` "
            "(or __graft_entry__.build())."
        )
    return ext


def use_hip(*tensors: torch.Tensor) -> bool:
    """True when the HIP kernel path should run for these tensors."""
    if force_eager():
        return False
    if not tensors or not tensors[0].is_cuda:
        return False
    return True


from .rmsnorm import rmsnorm  # noqa: E402
from .rope import (build_rope_cache, apply_rope, apply_rope_positions,  # noqa: E402
                   apply_rope_cs)
from .swiglu import swiglu  # noqa: E402
from .cross_entropy import shifted_cross_entropy  # noqa: E402
from .attention import causal_attention  # noqa: E402

__all__ = [
    "rmsnorm",
    "build_rope_cache",
    "apply_rope",
    "apply_rope_positions",
    "apply_rope_cs",
    "swiglu",
    "shifted_cross_entropy",
    "causal_attention",
    "extension",
    "use_hip",
    "force_eager",
]
