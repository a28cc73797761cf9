"""Seeding, distributed init, timers."""

from __future__ import annotations

import datetime
import logging
import os
import random
import time
from typing import Optional

import numpy as np
import torch
import torch.distributed as dist

logger = logging.getLogger(__name__)


def set_seed(seed: int, rank: int = 0) -> None:
    """Deterministic seeding (reference set_seed, trainer_base_ds_mp.py:124-129)."""
    random.seed(seed)
    np.random.seed(seed)
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(seed)


def get_rank() -> int:
    if dist.is_initialized():
        return dist.get_rank()
    return int(os.environ.get("RANK", 0))


def get_world_size() -> int:
    if dist.is_initialized():
        return dist.get_world_size()
    return int(os.environ.get("WORLD_SIZE", 1))


def init_distributed(backend: Optional[str] = None, timeout_s: int = 7200) -> tuple[int, int]:
    """init_process_group from torchrun env vars; nccl==RCCL on ROCm, same
    7200s timeout the reference passes (trainer_base_ds_mp.py:399).
    Returns (rank, world_size).  Single-process (no env) -> no init."""
    if dist.is_initialized():
        return dist.get_rank(), dist.get_world_size()
    world = int(os.environ.get("WORLD_SIZE", 1))
    if world <= 1:
        return 0, 1
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    if backend == "nccl" and torch.cuda.is_available():
        torch.cuda.set_device(local_rank)
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    dist.init_process_group(backend=backend, timeout=datetime.timedelta(seconds=timeout_s))
    return dist.get_rank(), dist.get_world_size()


import contextlib


@contextlib.contextmanager
def rank_zero_first(rank: int = None):
    """Rank-0-first execution discipline for dataset building/caching: rank 0
    runs the body while the others wait at a barrier, then the others run it
    (hitting the cache rank 0 produced) and everyone re-syncs — the
    reference's barrier pattern at trainer_base_ds_mp.py:163-176.  Barriers
    must be matched on ALL ranks, so call this on every rank."""
    if not dist.is_initialized():
        yield
        return
    if rank is None:
        rank = dist.get_rank()
    # barrier A: non-zero ranks wait in it while rank 0 runs the body;
    # rank 0 enters it after the body, releasing them.
    if rank != 0:
        dist.barrier()
    try:
        yield
    finally:
        if rank == 0:
            dist.barrier()  # barrier A (release)
        # barrier B: everyone re-syncs after their own pass over the body
        dist.barrier()


class StepTimer:
    """Cheap wall-clock step timer with device sync on CUDA."""

    def __init__(self, device: Optional[torch.device] = None):
        self.device = device
        self.history: list[float] = []
        self._t0: Optional[float] = None

    def start(self) -> None:
        if self.device is not None and self.device.type == "cuda":
            torch.cuda.synchronize(self.device)
        self._t0 = time.perf_counter()

    def stop(self) -> float:
        if self.device is not None and self.device.type == "cuda":
            torch.cuda.synchronize(self.device)
        dt = time.perf_counter() - self._t0
        self.history.append(dt)
        return dt

    @property
    def mean(self) -> float:
        return sum(self.history) / max(1, len(self.history))
