"""Lightweight observability: per-region GPU timers with achieved-
bandwidth reporting.

Analogue of the reference's event-based timing hooks (SURVEY §5;
reference elementwise.py:322-326 returns pyopencl events, and
test/common.py:41-56 prints ms + GB/s).  Here regions are bracketed
with HIP events through torch; every JIT/AOT kernel already carries a
distinguishable name for rocprofv3 kernel traces.

Usage::

    prof = Profiler(enabled=True)
    with prof.region("gradlap", bytes=nbytes):
        derivs(fx=f, lap=lap)
    print(prof.report())
"""

from __future__ import annotations

import time
from collections import defaultdict
from contextlib import contextmanager

import torch

__all__ = ["Profiler"]


class Profiler:
    def __init__(self, enabled=True, use_events=None):
        self.enabled = enabled
        self.use_events = (torch.cuda.is_available()
                           if use_events is None else use_events)
        self.times = defaultdict(float)
        self.calls = defaultdict(int)
        self.bytes = defaultdict(float)
        self._events = []

    @contextmanager
    def region(self, name, bytes=0):
        if not self.enabled:
            yield
            return
        if self.use_events:
            start = torch.cuda.Event(enable_timing=True)
            end = torch.cuda.Event(enable_timing=True)
            start.record()
            yield
            end.record()
            self._events.append((name, bytes, start, end))
        else:
            t0 = time.perf_counter()
            yield
            self.times[name] += time.perf_counter() - t0
            self.calls[name] += 1
            self.bytes[name] += bytes

    def _drain(self):
        if self._events:
            torch.cuda.synchronize()
            for name, nbytes, start, end in self._events:
                self.times[name] += start.elapsed_time(end) / 1e3
                self.calls[name] += 1
                self.bytes[name] += nbytes
            self._events.clear()

    def report(self):
        self._drain()
        lines = [f"{'region':24s} {'calls':>6s} {'total ms':>10s} "
                 f"{'ms/call':>9s} {'TB/s':>7s}"]
        for name in sorted(self.times, key=self.times.get, reverse=True):
            t = self.times[name]
            c = self.calls[name]
            b = self.bytes[name]
            bw = (b / t / 1e12) if t > 0 and b else 0.0
            lines.append(f"{name:24s} {c:6d} {t * 1e3:10.2f} "
                         f"{t / c * 1e3:9.3f} {bw:7.2f}")
        return "\n".join(lines)

    def reset(self):
        self._drain()
        self.times.clear()
        self.calls.clear()
        self.bytes.clear()
