"""Named timers with GPU-synchronized sections (reference training/timers.py)."""

from __future__ import annotations

import time
from collections import defaultdict
from contextlib import contextmanager

import torch


class Timers:
    def __init__(self, cuda_sync: bool = True):
        self.cuda_sync = cuda_sync and torch.cuda.is_available()
        self.totals: dict[str, float] = defaultdict(float)
        self.counts: dict[str, int] = defaultdict(int)

    @contextmanager
    def __call__(self, name: str):
        if self.cuda_sync:
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        try:
            yield
        finally:
            if self.cuda_sync:
                torch.cuda.synchronize()
            self.totals[name] += time.perf_counter() - t0
            self.counts[name] += 1

    def mean(self, name: str) -> float:
        return self.totals[name] / max(1, self.counts[name])

    def summary(self) -> dict[str, float]:
        return {k: self.mean(k) for k in self.totals}

    def reset(self) -> None:
        self.totals.clear()
        self.counts.clear()
