"""Small utilities: seeding and HIP-event step timing."""
from __future__ import annotations

import random
import time
from typing import List

import numpy as np
import torch


def seed_everything(seed: int):
    random.seed(seed)
    np.random.seed(seed)
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(seed)


class StepTimer:
    """Per-step wall timing; uses HIP events on GPU (torch.cuda.Event is a
    hipEvent on ROCm) so device work is measured, not launch time."""

    def __init__(self, device: torch.device):
        self.device = device
        self.use_events = device.type == "cuda"
        self.times_ms: List[float] = []
        self._start = None

    def start(self):
        if self.use_events:
            self._start = torch.cuda.Event(enable_timing=True)
            self._end = torch.cuda.Event(enable_timing=True)
            self._start.record()
        else:
            self._t0 = time.perf_counter()

    def stop(self):
        if self.use_events:
            self._end.record()
            self._end.synchronize()
            self.times_ms.append(self._start.elapsed_time(self._end))
        else:
            self.times_ms.append((time.perf_counter() - self._t0) * 1e3)

    def mean_ms(self) -> float:
        return sum(self.times_ms) / max(len(self.times_ms), 1)
