"""Tracing helpers (SURVEY.md §5 'Tracing / profiling').

The reference has only wall-clock wps prints; here:
  * `range(name)` emits roctx ranges (visible in rocprofv3 --sys-trace
    timelines) when ZAREMBA_AMD_ROCTX=1 and roctx is available — used by
    the trainer around forward / loss / backward / step,
  * `StepTimer` gives cheap CUDA-event step timing for ad-hoc profiling
    without a profiler attached.
"""

from __future__ import annotations

import contextlib
import os

import torch

_roctx = None
if os.environ.get("ZAREMBA_AMD_ROCTX", "0") == "1":
    try:
        from torch.cuda import nvtx as _roctx  # maps to roctx on ROCm
    except Exception:  # pragma: no cover
        _roctx = None


@contextlib.contextmanager
def trace_range(name: str):
    if _roctx is not None:
        _roctx.range_push(name)
        try:
            yield
        finally:
            _roctx.range_pop()
    else:
        yield


class StepTimer:
    """CUDA-event based timer: collects per-step milliseconds."""

    def __init__(self, enabled: bool = True):
        self.enabled = enabled and torch.cuda.is_available()
        self.times = []
        self._start = None

    def start(self):
        if not self.enabled:
            return
        self._start = torch.cuda.Event(enable_timing=True)
        self._end = torch.cuda.Event(enable_timing=True)
        self._start.record()

    def stop(self):
        if not self.enabled or self._start is None:
            return
        self._end.record()
        self._end.synchronize()
        self.times.append(self._start.elapsed_time(self._end))
        self._start = None

    def summary(self):
        if not self.times:
            return {}
        t = sorted(self.times)
        return {
            "n": len(t),
            "mean_ms": sum(t) / len(t),
            "p50_ms": t[len(t) // 2],
            "max_ms": t[-1],
        }
