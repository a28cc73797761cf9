"""Device-true section timers (hipEvent pairs via torch.cuda.Event).

Round-1's host-side timers bracketed kernel ENQUEUE, not execution — with
async launches the forward timer measured microseconds while all the work
drained inside the first synchronizing call (the backward), misattributing
~everything to "backward".  These timers record an event pair on the
CURRENT stream around each section; elapsed_time between the events is
device-measured wall time of the stream segment, which is the truth for
fwd/bwd compute and, around a comm wait, the stall the compute stream
actually suffered.

CPU fallback: perf_counter pairs (exact there — CPU ops are synchronous).

Usage:
    timers = DeviceTimers(device)
    with timers.section("forward"):
        ...
    totals = timers.summary(reset=True)   # {"forward": seconds, ...}

``summary`` synchronizes once and folds all outstanding event pairs.
"""

from __future__ import annotations

import contextlib
import time
from collections import defaultdict
from typing import Dict, List, Optional

import torch


class DeviceTimers:
    def __init__(self, device: Optional[torch.device] = None, enabled: bool = True):
        self.device = device
        self.enabled = enabled
        self.use_events = bool(
            device is not None and device.type == "cuda" and torch.cuda.is_available()
        )
        self._totals: Dict[str, float] = defaultdict(float)
        self._pairs: Dict[str, List] = defaultdict(list)  # (start_ev, end_ev)
        self._pool: List = []  # recycled events

    # ------------------------------------------------------------------
    def _event(self):
        if self._pool:
            return self._pool.pop()
        return torch.cuda.Event(enable_timing=True)

    @contextlib.contextmanager
    def section(self, name: str):
        if not self.enabled:
            yield
            return
        if self.use_events:
            start = self._event()
            start.record()
            try:
                yield
            finally:
                end = self._event()
                end.record()
                self._pairs[name].append((start, end))
        else:
            t0 = time.perf_counter()
            try:
                yield
            finally:
                self._totals[name] += time.perf_counter() - t0

    def _fold(self) -> None:
        if not self.use_events:
            return
        any_pairs = any(self._pairs.values())
        if not any_pairs:
            return
        torch.cuda.synchronize(self.device)
        for name, pairs in self._pairs.items():
            for start, end in pairs:
                self._totals[name] += start.elapsed_time(end) / 1000.0
                self._pool.append(start)
                self._pool.append(end)
            pairs.clear()

    # ------------------------------------------------------------------
    def summary(self, reset: bool = True) -> Dict[str, float]:
        """Fold outstanding event pairs (synchronizes on GPU) and return the
        accumulated seconds per section."""
        self._fold()
        out = dict(self._totals)
        if reset:
            self._totals.clear()
        return out

    def totals_nosync(self) -> Dict[str, float]:
        """Already-folded totals without synchronizing (CPU path: complete)."""
        return dict(self._totals)
