"""rocprof-visible tracing ranges.

The reference had only per-round wall-clock timing (src/blades/simulator.py:453-455).
Here every phase of the round schedule is wrapped in a named range so
``rocprofv3 --sys-trace`` / roctx shows broadcast / local-train / gather /
attack / aggregate / apply as separate regions.  Falls back to a no-op when
no GPU (or no roctx) is present so CPU CI never depends on ROCm markers.
"""
from __future__ import annotations

import contextlib
import time
from collections import defaultdict

import torch

_HAS_NVTX = torch.cuda.is_available()

# Cumulative host-side phase timers (always on; negligible overhead).
phase_seconds = defaultdict(float)


@contextlib.contextmanager
def trace_range(name: str):
    t0 = time.perf_counter()
    if _HAS_NVTX:
        torch.cuda.nvtx.range_push(name)  # roctx range on ROCm builds
    try:
        yield
    finally:
        if _HAS_NVTX:
            torch.cuda.nvtx.range_pop()
        phase_seconds[name] += time.perf_counter() - t0


def annotate(name: str):
    """Decorator form of :func:`trace_range`."""

    def deco(fn):
        def wrapped(*a, **kw):
            with trace_range(name):
                return fn(*a, **kw)

        wrapped.__name__ = fn.__name__
        return wrapped

    return deco
