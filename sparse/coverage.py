"""Provenance tracking (reference sparse/coverage.py:50-109).

Public API entry points are wrapped so profiles attribute GPU work to user
code: each call pushes a roctx/torch.profiler range named after the API
function.  Ranges are emitted only while profiling is enabled
(sparse.profiling.enable()) so the hot path stays free of overhead.
"""
from __future__ import annotations

from functools import wraps
from types import FunctionType, MethodType, ModuleType
from typing import Callable

_PROFILING = False
_STACK = []


def profiling_enabled() -> bool:
    return _PROFILING


def enable_profiling(flag: bool = True) -> None:
    global _PROFILING
    _PROFILING = flag


def current_provenance() -> str:
    return _STACK[-1] if _STACK else ""


def track_provenance(nested: bool = False) -> Callable:
    def deco(func):
        name = getattr(func, "__qualname__", getattr(func, "__name__", "op"))

        @wraps(func)
        def wrapper(*args, **kwargs):
            if not _PROFILING:
                return func(*args, **kwargs)
            import torch

            _STACK.append(name)
            try:
                with torch.profiler.record_function(f"sparse::{name}"):
                    return func(*args, **kwargs)
            finally:
                _STACK.pop()

        return wrapper

    return deco


def should_wrap(obj: object) -> bool:
    return isinstance(obj, (FunctionType, MethodType))


def clone_module(origin_module: ModuleType, new_globals: dict) -> None:
    """Wrap functions that shadow names from origin_module with provenance
    tracking (reference coverage.py:58-86)."""
    for attr, value in list(new_globals.items()):
        if attr not in origin_module.__dict__:
            continue
        if isinstance(value, FunctionType):
            new_globals[attr] = track_provenance(nested=True)(value)


def clone_scipy_arr_kind(origin_class: type) -> Callable[[type], type]:
    """Class decorator: wrap methods that exist on the scipy counterpart
    (reference coverage.py:88-109)."""

    def body(cls: type) -> type:
        for attr, value in list(cls.__dict__.items()):
            if not hasattr(origin_class, attr):
                continue
            if should_wrap(value):
                setattr(cls, attr, track_provenance(nested=True)(value))
        return cls

    return body
