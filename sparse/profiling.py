"""Profiling / tracing helpers (SURVEY §5 aux subsystem).

The reference leans on Legion's profiler plus provenance tags
(coverage.py:50-109) and timing futures (examples/benchmark.py:18-31); here:

- enable()/disable(): turn on provenance ranges (sparse.coverage) so
  torch.profiler / rocprofv3 attribute GPU work to user-level API calls.
- profile(path): context manager running torch.profiler (CPU+GPU) and
  writing a chrome trace; provenance ranges show as `sparse::<api>`.
- timer(): a HIP-event timer that does not block until stop (the
  legate.timing futures analog).
"""
from __future__ import annotations

import contextlib
from typing import Optional

from . import coverage

enable = coverage.enable_profiling


def disable() -> None:
    coverage.enable_profiling(False)


@contextlib.contextmanager
def profile(trace_path: Optional[str] = None, activities=None):
    import torch

    coverage.enable_profiling(True)
    acts = activities
    if acts is None:
        acts = [torch.profiler.ProfilerActivity.CPU]
        if torch.cuda.is_available():
            acts.append(torch.profiler.ProfilerActivity.CUDA)
    prof = torch.profiler.profile(activities=acts)
    prof.__enter__()
    try:
        yield prof
    finally:
        prof.__exit__(None, None, None)
        coverage.enable_profiling(False)
        if trace_path:
            prof.export_chrome_trace(trace_path)


class timer:
    """start()/stop() -> ms; HIP events on GPU, perf_counter on CPU."""

    def __init__(self):
        import torch

        self._gpu = torch.cuda.is_available()
        self._t0 = None

    def start(self):
        if self._gpu:
            import torch

            self._t0 = torch.cuda.Event(enable_timing=True)
            self._t0.record()
        else:
            from time import perf_counter_ns

            self._t0 = perf_counter_ns()

    def stop(self) -> float:
        if self._gpu:
            import torch

            end = torch.cuda.Event(enable_timing=True)
            end.record()
            end.synchronize()
            return self._t0.elapsed_time(end)
        from time import perf_counter_ns

        return (perf_counter_ns() - self._t0) / 1e6
