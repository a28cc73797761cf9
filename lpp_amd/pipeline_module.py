"""PipelineModule: partition the flat LayerSpec list, build only this rank's
stage, expose the grid.

Native replacement for DeepSpeed's ``PipelineModule`` as the reference uses
it (trainer_base_ds_mp.py:33,425-429): LayerSpec lazy build, param-balanced
contiguous partition, per-stage module materialisation, topology queries.
"""

from __future__ import annotations

from typing import Callable, List, Optional, Sequence

import torch
import torch.nn as nn

from .layer_spec import LayerSpec, partition_balanced, partition_uniform
from .topology import ProcessGrid


class PipelineModule(nn.Module):
    def __init__(
        self,
        layers: Sequence[LayerSpec],
        grid: ProcessGrid,
        loss_fn: Optional[Callable] = None,
        partition_method: str = "parameters",
        activation_checkpoint_interval: int = 0,
        device: Optional[torch.device] = None,
        dtype: Optional[torch.dtype] = None,
    ):
        super().__init__()
        self.specs = list(layers)
        self.grid = grid
        self.loss_fn = loss_fn
        self.activation_checkpoint_interval = activation_checkpoint_interval

        if partition_method == "parameters":
            def _w(s):
                if isinstance(s, LayerSpec):
                    return max(1, s.param_count())
                if isinstance(s, nn.Module):  # family-A raw modules
                    return max(1, sum(p.numel() for p in s.parameters()))
                return 1

            weights = [_w(s) for s in self.specs]
            self.bounds = partition_balanced(weights, grid.num_stages)
        elif partition_method == "uniform":
            self.bounds = partition_uniform(len(self.specs), grid.num_stages)
        else:
            raise ValueError(f"unknown partition_method {partition_method!r}")

        s = grid.stage_id
        self.local_start = self.bounds[s]
        self.local_stop = self.bounds[s + 1]

        built: List[nn.Module] = []
        prev_dtype = torch.get_default_dtype()
        try:
            if dtype is not None:
                torch.set_default_dtype(dtype)
            import contextlib

            ctx = torch.device(device) if device is not None else contextlib.nullcontext()
            with ctx:
                for spec in self.specs[self.local_start : self.local_stop]:
                    m = spec.build() if isinstance(spec, LayerSpec) else spec
                    built.append(m)
        finally:
            torch.set_default_dtype(prev_dtype)
        self.layers = nn.ModuleList(built)
        if device is not None or dtype is not None:
            self.layers.to(device=device, dtype=dtype)
        # Family-A (raw-module specs): drop references to OFF-STAGE modules so
        # the monolithic source model's other-stage weights become collectable
        # once the caller releases its own handle.  The reference materialises
        # the full model per worker and keeps it (README.md:21); we keep only
        # this stage's slice.  LayerSpec entries are weightless and stay.
        self.num_specs = len(self.specs)
        for i, s in enumerate(self.specs):
            if isinstance(s, nn.Module) and not (self.local_start <= i < self.local_stop):
                self.specs[i] = None

    # -- queries -----------------------------------------------------------
    @property
    def num_local_layers(self) -> int:
        return len(self.layers)

    def global_layer_index(self, local_idx: int) -> int:
        """Local layer i -> index in the flat spec list == checkpoint layer
        file number (convert2ckpt.py:23-36)."""
        return self.local_start + local_idx

    def stage_of_layer(self, global_idx: int) -> int:
        for s in range(self.grid.num_stages):
            if self.bounds[s] <= global_idx < self.bounds[s + 1]:
                return s
        raise IndexError(global_idx)

    # -- compute -----------------------------------------------------------
    def forward(self, x: torch.Tensor) -> torch.Tensor:
        """Run the local stage.  ``activation_checkpoint_interval`` groups
        layers and recomputes each group in backward (DeepSpeed semantics —
        trainer_base_ds_mp.py:428; interval 0 disables engine-level
        checkpointing, leaving any per-layer flag to the layer itself)."""
        interval = self.activation_checkpoint_interval
        if interval > 0 and self.training and torch.is_grad_enabled():
            i = 0
            n = len(self.layers)
            while i < n:
                group = self.layers[i : i + interval]
                if any(p.requires_grad for layer in group for p in layer.parameters()):
                    x = torch.utils.checkpoint.checkpoint(
                        self._run_group, i, i + len(group), x,
                        use_reentrant=False, preserve_rng_state=False,
                    )
                else:
                    x = self._run_group(i, i + len(group), x)
                i += len(group)
            return x
        for layer in self.layers:
            x = layer(x)
        return x

    def _run_group(self, start: int, stop: int, x: torch.Tensor) -> torch.Tensor:
        for layer in self.layers[start:stop]:
            x = layer(x)
        return x

    def extra_repr(self) -> str:
        return (
            f"stage={self.grid.stage_id}/{self.grid.num_stages} "
            f"layers=[{self.local_start},{self.local_stop}) of {self.num_specs}"
        )
