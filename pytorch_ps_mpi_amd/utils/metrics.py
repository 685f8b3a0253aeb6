"""Per-step metrics — the successor of the reference's timing dict
(ps.py:116,135-191): `step()` returns ``(loss, metrics)`` where metrics is a
plain dict of accumulated spans and byte counters for the step.

Two layers:
  * host wall-clock around enqueue/wait points (always on, no stream syncs);
  * optional HIP-event spans (``PS(..., profile_gpu=True)``): each timer
    also records start/end events on the current stream and `finalize_gpu`
    resolves them into ``<key>_gpu_ms`` entries (costs one synchronize per
    step — a profiling mode, like the reference's per-step instrumentation).
Kernel-level detail comes from rocprofv3, not from here.
"""

from __future__ import annotations

import time

import torch


class StepMetrics(dict):
    def __init__(self, gpu=False):
        super().__init__()
        self._gpu = bool(gpu) and torch.cuda.is_available()
        self._spans = []

    def add(self, key, val):
        self[key] = self.get(key, 0.0) + val

    def timer(self, key):
        return _Span(self, key)

    def finalize_gpu(self):
        if not self._spans:
            return
        torch.cuda.synchronize()
        for key, s, e in self._spans:
            self.add(key + "_gpu_ms", s.elapsed_time(e))
        self._spans = []


class _Span:
    __slots__ = ("m", "key", "t0", "ev")

    def __init__(self, m, key):
        self.m = m
        self.key = key
        self.ev = None

    def __enter__(self):
        if self.m._gpu:
            s = torch.cuda.Event(enable_timing=True)
            s.record()
            self.ev = s
        self.t0 = time.perf_counter()
        return self

    def __exit__(self, *a):
        self.m.add(self.key, time.perf_counter() - self.t0)
        if self.ev is not None:
            e = torch.cuda.Event(enable_timing=True)
            e.record()
            self.m._spans.append((self.key, self.ev, e))
        return False


def print_summary(metrics_list, file=None):
    """Aggregate and pretty-print a list of per-step metric dicts — the
    moral successor of the reference's print_summary (mpi_comms.py:176-184).
    """
    import sys
    out = file or sys.stdout
    keys = sorted({k for m in metrics_list for k in m
                   if isinstance(m.get(k), (int, float))})
    n = max(1, len(metrics_list))
    print(f"{'metric':28s} {'mean':>12s} {'max':>12s}", file=out)
    for k in keys:
        vals = [m[k] for m in metrics_list if k in m]
        mean = sum(vals) / len(vals)
        print(f"{k:28s} {mean:12.6g} {max(vals):12.6g}", file=out)
