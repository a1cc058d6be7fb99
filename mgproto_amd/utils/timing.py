"""Per-phase step timing (observability; SURVEY.md §5 tracing).

CUDA-event based phase timers with negligible overhead when disabled
(wall-clock fallback on CPU). Enable with ``MGPROTO_TIMING=1``; the
trainer then logs per-phase means every print interval. For kernel-level
data use rocprofv3 (profiles/).
"""

import os
import time
from collections import defaultdict
from contextlib import contextmanager

import torch


class PhaseTimer:
    def __init__(self, enabled=None, device=None):
        if enabled is None:
            enabled = os.environ.get('MGPROTO_TIMING') == '1'
        self.enabled = enabled
        self.use_events = enabled and torch.cuda.is_available()
        self._events = []          # (name, start_ev, end_ev)
        self.sums = defaultdict(float)
        self.counts = defaultdict(int)

    @contextmanager
    def phase(self, name):
        if not self.enabled:
            yield
            return
        if not self.use_events:            # CPU: wall clock
            t0 = time.perf_counter()
            try:
                yield
            finally:
                self.sums[name] += (time.perf_counter() - t0) * 1e3
                self.counts[name] += 1
            return
        s = torch.cuda.Event(enable_timing=True)
        e = torch.cuda.Event(enable_timing=True)
        s.record()
        try:
            yield
        finally:
            e.record()
            self._events.append((name, s, e))

    def collect(self):
        """Resolve finished events (synchronizes on the newest one)."""
        if not self._events:
            return
        self._events[-1][2].synchronize()
        for name, s, e in self._events:
            self.sums[name] += s.elapsed_time(e)
            self.counts[name] += 1
        self._events.clear()

    def summary(self, reset=True):
        self.collect()
        out = {name: self.sums[name] / max(self.counts[name], 1)
               for name in self.sums}
        if reset:
            self.sums.clear()
            self.counts.clear()
        return out
