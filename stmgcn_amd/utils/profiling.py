"""Step-level observability (SURVEY §5 tracing/profiling).

The reference has only wall-clock prints (Model_Trainer.py:21,62); here:
  - StepTimer: HIP-event per-step timing with negligible overhead, usable
    inside the training loop (GPU) or falling back to perf_counter (CPU)
  - kernel-level profiling is external by design: the rocprofv3 recipe that
    produced profiles/ is

      rocprofv3 --kernel-trace --stats -d OUT -o stats -- python bench.py ...
      rocprofv3 --pmc MfmaUtil LdsBankConflict MeanOccupancyPerActiveCU \
          -d OUT -o pmc -- python bench/kernel_micro.py

    (PMC counters must be collected in their own run, never combined with
    the trace domains.)
"""
from __future__ import annotations

import json
import time
from typing import List

import torch


class StepTimer:
    """Ring of HIP event pairs; `with timer: step()` then `timer.ms()`.

    Events are recorded asynchronously; ms() synchronizes only the events it
    reads, so timing N steps costs two event records per step.
    """

    def __init__(self, capacity: int = 256):
        self.capacity = capacity
        self._gpu = torch.cuda.is_available()
        self._events: List = []
        self._cpu_times: List[float] = []
        self._t0 = 0.0

    def __enter__(self):
        if self._gpu:
            s = torch.cuda.Event(enable_timing=True)
            s.record()
            self._start = s
        else:
            self._t0 = time.perf_counter()
        return self

    def __exit__(self, *exc):
        if self._gpu:
            e = torch.cuda.Event(enable_timing=True)
            e.record()
            self._events.append((self._start, e))
            if len(self._events) > self.capacity:
                self._events.pop(0)
        else:
            self._cpu_times.append((time.perf_counter() - self._t0) * 1e3)
            if len(self._cpu_times) > self.capacity:
                self._cpu_times.pop(0)
        return False

    def ms(self) -> List[float]:
        if self._gpu:
            if self._events:
                self._events[-1][1].synchronize()
            return [s.elapsed_time(e) for s, e in self._events]
        return list(self._cpu_times)

    def summary(self) -> dict:
        xs = sorted(self.ms())
        if not xs:
            return {"n": 0}
        return {
            "n": len(xs),
            "mean_ms": sum(xs) / len(xs),
            "p50_ms": xs[len(xs) // 2],
            "p95_ms": xs[min(len(xs) - 1, int(len(xs) * 0.95))],
            "min_ms": xs[0],
            "max_ms": xs[-1],
        }

    def dump(self, path: str):
        with open(path, "a") as f:
            f.write(json.dumps(self.summary()) + "\n")
