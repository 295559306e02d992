"""Per-round phase timing (the reference only had ad-hoc
wall-clock counters inside aggregators; here the round loop itself reports a
train / exchange / aggregate / evaluate breakdown).

Timing is OFF by default because accurate GPU phase times need a
``torch.cuda.synchronize`` at each phase boundary, which breaks the async
round pipeline. Enable with MURMURA_TIMING=1 (bench/diagnostics)."""

from __future__ import annotations

import os
import time
from collections import defaultdict
from typing import Dict

import torch


def timing_enabled() -> bool:
    return os.environ.get("MURMURA_TIMING") == "1"


class PhaseTimer:
    def __init__(self, device=None):
        self.enabled = timing_enabled()
        self.device = device
        self.totals: Dict[str, float] = defaultdict(float)
        self.counts: Dict[str, int] = defaultdict(int)
        self._t0 = None

    def _sync(self):
        if self.device is not None and self.device.type == "cuda":
            torch.cuda.synchronize(self.device)

    def phase(self, name: str):
        return _Phase(self, name)

    def summary(self) -> Dict[str, float]:
        """Mean milliseconds per phase."""
        return {
            k: self.totals[k] / max(1, self.counts[k]) * 1000.0 for k in self.totals
        }


class _Phase:
    def __init__(self, timer: PhaseTimer, name: str):
        self.timer = timer
        self.name = name

    def __enter__(self):
        if self.timer.enabled:
            self.timer._sync()
            self._t0 = time.perf_counter()
        return self

    def __exit__(self, *exc):
        if self.timer.enabled:
            self.timer._sync()
            self.timer.totals[self.name] += time.perf_counter() - self._t0
            self.timer.counts[self.name] += 1
        return False
