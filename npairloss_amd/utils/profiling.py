"""Per-phase timing instrumentation (the tracing/observability subsystem
the reference lacked — SURVEY.md section 5).

PhaseTimers wraps training-step phases (data/forward/loss/backward/comm/
optimizer) in CUDA/HIP events so GPU time is measured without host syncs
until `report()`; on CPU it falls back to wall clock.  Used by bench.py
--timers and available to any Trainer user.
"""

from __future__ import annotations

import collections
import time
from contextlib import contextmanager
from typing import Dict, List, Tuple

import torch


class PhaseTimers:
    def __init__(self, enabled: bool = True, use_cuda: bool = None):
        self.enabled = enabled
        self.use_cuda = torch.cuda.is_available() if use_cuda is None else use_cuda
        self._events: List[Tuple[str, object, object]] = []
        self._cpu_acc: Dict[str, float] = collections.defaultdict(float)
        self._counts: Dict[str, int] = collections.defaultdict(int)

    @contextmanager
    def phase(self, name: str):
        if not self.enabled:
            yield
            return
        if self.use_cuda:
            e0 = torch.cuda.Event(enable_timing=True)
            e1 = torch.cuda.Event(enable_timing=True)
            e0.record()
            try:
                yield
            finally:
                e1.record()
                self._events.append((name, e0, e1))
                self._counts[name] += 1
        else:
            t0 = time.perf_counter()
            try:
                yield
            finally:
                self._cpu_acc[name] += time.perf_counter() - t0
                self._counts[name] += 1

    def report(self, reset: bool = True) -> Dict[str, float]:
        """Total milliseconds per phase (synchronizes once on GPU)."""
        out: Dict[str, float] = collections.defaultdict(float)
        if self.use_cuda and self._events:
            torch.cuda.synchronize()
            for name, e0, e1 in self._events:
                out[name] += e0.elapsed_time(e1)
        for name, sec in self._cpu_acc.items():
            out[name] += sec * 1000.0
        result = dict(out)
        if reset:
            self._events.clear()
            self._cpu_acc.clear()
            self._counts.clear()
        return result
