"""Shared example/benchmark harness (role of reference examples/benchmark.py).

Timer protocol: start()/stop() -> ms, non-blocking until stop (HIP events on
GPU, reference's legate.timing futures parity).  parse_common_args selects
the backing package: "sparse" (this framework) or "scipy" (oracle).
"""
from __future__ import annotations

import argparse
from typing import Protocol


class Timer(Protocol):
    def start(self) -> None: ...

    def stop(self) -> float: ...


class HipTimer:
    """torch.cuda (HIP) event timer — doesn't block until stop()."""

    def __init__(self):
        self._ev = None

    def start(self):
        import torch

        self._ev = torch.cuda.Event(enable_timing=True)
        self._ev.record()

    def stop(self):
        import torch
        import torch.distributed as dist

        end = torch.cuda.Event(enable_timing=True)
        end.record()
        if dist.is_available() and dist.is_initialized():
            dist.barrier()
        end.synchronize()
        ms = self._ev.elapsed_time(end)
        return ms


class CpuTimer:
    def __init__(self):
        self._t = None

    def start(self):
        from time import perf_counter_ns

        self._t = perf_counter_ns()

    def stop(self):
        from time import perf_counter_ns
        import torch.distributed as dist

        if dist.is_available() and dist.is_initialized():
            dist.barrier()
        return (perf_counter_ns() - self._t) / 1e6


class DummyScope:
    def __enter__(self):
        return self

    def __exit__(self, *a):
        return None

    def __getitem__(self, item):
        return self

    def count(self, *_):
        return 1


def get_phase_procs(use_sparse: bool):
    return DummyScope(), DummyScope()


def parse_common_args(argv=None):
    parser = argparse.ArgumentParser(add_help=False)
    parser.add_argument("--package", default="sparse",
                        choices=["sparse", "legate", "scipy"])
    args, _ = parser.parse_known_args(argv)
    if args.package in ("sparse", "legate"):
        import sparse
        import sparse.linalg as linalg
        from sparse import darray as np_like

        rt = sparse.runtime()
        timer = HipTimer() if rt.use_gpu else CpuTimer()
        return args.package, timer, np_like, sparse, linalg, True
    else:
        import numpy as np_like
        import scipy.sparse as sparse_mod
        import scipy.sparse.linalg as linalg

        return args.package, CpuTimer(), np_like, sparse_mod, linalg, False
