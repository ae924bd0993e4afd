"""Tracing hooks (SURVEY.md §5.1 rebuild note).

The reference had no tracing of its own (Chainer TimerHook idioms).  Here:
``trace_range('name')`` emits roctx ranges (via torch.cuda.nvtx, which maps
to roctracer markers on ROCm) so `rocprofv3 --kernel-trace` runs show the
framework phases (pack / all-reduce / update / bn ...), plus a
``StepTimer`` that aggregates wall / compute / comm time per step.

Debug environment knobs:
  AMD_SERIALIZE_KERNEL=3  AMD_SERIALIZE_COPY=3   serialize for race hunts
  RCCL_DEBUG=INFO                                collective tracing
"""

import contextlib
import time

import torch

_NVTX = torch.cuda.is_available()


@contextlib.contextmanager
def trace_range(name):
    if _NVTX:
        torch.cuda.nvtx.range_push(name)
    try:
        yield
    finally:
        if _NVTX:
            torch.cuda.nvtx.range_pop()


class StepTimer(object):
    """Wall-clock per-step phase timer; aggregates into a dict."""

    def __init__(self):
        self.totals = {}
        self.counts = {}

    @contextlib.contextmanager
    def phase(self, name, sync=False):
        if sync and torch.cuda.is_available():
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        with trace_range(name):
            yield
        if sync and torch.cuda.is_available():
            torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        self.totals[name] = self.totals.get(name, 0.0) + dt
        self.counts[name] = self.counts.get(name, 0) + 1

    def summary(self):
        return {k: {'total_s': v, 'mean_ms': v / self.counts[k] * 1000}
                for k, v in self.totals.items()}
