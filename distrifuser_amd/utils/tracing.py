"""Lightweight tracing / observability (SURVEY §5 aux subsystems).

* ``trace_range(name)`` — roctx ranges via torch.cuda.nvtx (maps to roctx on
  ROCm, visible in rocprofv3 --sys-trace timelines), enabled with DFA_TRACE=1.
* ``StepTimer`` — per-denoise-step wall/device timing ring buffer.
The comm manager keeps its own per-collective byte/launch counters
(utils/comm.py stats).
"""

from __future__ import annotations

import contextlib
import os
import time

import torch

TRACE = os.environ.get("DFA_TRACE", "0") == "1"


@contextlib.contextmanager
def trace_range(name: str):
    if TRACE and torch.cuda.is_available():
        torch.cuda.nvtx.range_push(name)
        try:
            yield
        finally:
            torch.cuda.nvtx.range_pop()
    else:
        yield


class StepTimer:
    """Wall-clock per-step timer with a device sync per sample (debug aid)."""

    def __init__(self, enabled: bool = False, sync: bool = True):
        self.enabled = enabled
        self.sync = sync
        self.samples: list[float] = []
        self._t0: float | None = None

    def start(self) -> None:
        if not self.enabled:
            return
        if self.sync and torch.cuda.is_available():
            torch.cuda.synchronize()
        self._t0 = time.perf_counter()

    def stop(self) -> None:
        if not self.enabled or self._t0 is None:
            return
        if self.sync and torch.cuda.is_available():
            torch.cuda.synchronize()
        self.samples.append(time.perf_counter() - self._t0)
        self._t0 = None

    def summary(self) -> dict:
        if not self.samples:
            return {}
        s = sorted(self.samples)
        return {
            "n": len(s),
            "mean_ms": sum(s) / len(s) * 1e3,
            "p50_ms": s[len(s) // 2] * 1e3,
            "max_ms": s[-1] * 1e3,
        }
