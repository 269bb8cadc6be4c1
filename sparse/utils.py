"""Helpers (reference sparse/utils.py).

- cast_to_common_type (utils.py:120-140)
- factor_int near-square grid (utils.py:144-150)
- tensor/DistArray interop (the store<->cunumeric helpers, utils.py:46-91,
  map onto torch<->DistArray here)
"""
from __future__ import annotations

import math
import warnings
from typing import Tuple

import numpy as np
import torch

from .darray import DistArray, asdistarray
from .types import common_value_dtype, to_torch_dtype


def cast_to_common_type(*arrays):
    """Promote all operands to one value dtype (reference utils.py:120-140)."""
    dts = []
    for a in arrays:
        if isinstance(a, torch.Tensor):
            dts.append(a.dtype)
        elif isinstance(a, DistArray):
            dts.append(a.tdtype)
        elif isinstance(a, np.ndarray):
            dts.append(to_torch_dtype(a.dtype))
        else:
            dts.append(to_torch_dtype(a.dtype))
    target = common_value_dtype(*dts)
    out = []
    for a in arrays:
        if isinstance(a, DistArray):
            out.append(a.astype(target) if a.tdtype != target else a)
        elif isinstance(a, torch.Tensor):
            out.append(a.to(target))
        elif isinstance(a, np.ndarray):
            out.append(asdistarray(a).astype(target))
        else:
            out.append(a.astype(target))
    return tuple(out)


def factor_int(num: int) -> Tuple[int, int]:
    """Near-square factorization for 2-D grids (reference utils.py:144-150)."""
    for a in range(int(math.isqrt(num)), 0, -1):
        if num % a == 0:
            return (a, num // a)
    return (1, num)


def get_tensor_from_distarray(x: DistArray) -> torch.Tensor:
    """Global tensor, replicated (the store_to_cunumeric_array analog)."""
    return x.gather()


def distarray_from_tensor(t: torch.Tensor) -> DistArray:
    return DistArray.from_global(t)


def perf_warning(msg: str) -> None:
    """User-facing performance hazard warning (reference utils.py:31-37)."""
    warnings.warn(msg, UserWarning, stacklevel=3)
