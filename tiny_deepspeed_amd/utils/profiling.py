"""Lightweight observability helpers (the reference has none — SURVEY.md 5.1).

- roctx ranges: torch.cuda.nvtx maps to rocTX on ROCm, so rocprofv3's
  marker trace shows named phases (forward/backward/step/comm).
- StepTimer: hipEvent-based per-step timing with no host syncs until read.
Enabled via TDSA_TRACE=1 (ranges are no-ops otherwise: zero overhead).
"""

import contextlib
import os
import time

import torch

_TRACE = os.environ.get("TDSA_TRACE", "0") == "1"


@contextlib.contextmanager
def trace_range(name):
    if _TRACE and torch.cuda.is_available():
        torch.cuda.nvtx.range_push(name)
        try:
            yield
        finally:
            torch.cuda.nvtx.range_pop()
    else:
        yield


class StepTimer:
    """Times device work between start()/stop() with hipEvents on GPU,
    wall-clock on CPU. Read elapsed_ms() after the fact."""

    def __init__(self):
        self.is_cuda = torch.cuda.is_available()
        self._evs = []
        self._t0 = None
        self._ms = []

    def start(self):
        if self.is_cuda:
            e = torch.cuda.Event(enable_timing=True)
            e.record()
            self._evs.append([e, None])
        else:
            self._t0 = time.perf_counter()

    def stop(self):
        if self.is_cuda:
            e = torch.cuda.Event(enable_timing=True)
            e.record()
            self._evs[-1][1] = e
        else:
            self._ms.append((time.perf_counter() - self._t0) * 1e3)

    def elapsed_ms(self):
        if self.is_cuda:
            torch.cuda.synchronize()
            return [a.elapsed_time(b) for a, b in self._evs if b is not None]
        return list(self._ms)
