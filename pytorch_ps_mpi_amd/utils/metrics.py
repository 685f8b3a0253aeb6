"""Per-step metrics — the successor of the reference's timing dict
(ps.py:116,135-191): `step()` returns ``(loss, metrics)`` where metrics is a
plain dict of accumulated spans and byte counters for the step.

Timers are host wall-clock around enqueue/wait points (cheap, no stream
syncs); kernel-accurate timing comes from rocprofv3, not from here.
"""

from __future__ import annotations

import time


class StepMetrics(dict):
    def add(self, key, val):
        self[key] = self.get(key, 0.0) + val

    def timer(self, key):
        return _Span(self, key)


class _Span:
    __slots__ = ("m", "key", "t0")

    def __init__(self, m, key):
        self.m = m
        self.key = key

    def __enter__(self):
        self.t0 = time.perf_counter()
        return self

    def __exit__(self, *a):
        self.m.add(self.key, time.perf_counter() - self.t0)
        return False
