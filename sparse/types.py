"""Type maps shared across the package.

Reference parity: sparse/types.py:20-21 (coord_ty=int64, nnz_ty=uint64) and
the value-dtype set of src/sparse/util/dispatch.h:27-74
(float32/float64/complex64/complex128 x {int32,int64} indices).

MI355X-native choice: indices are stored as int32 whenever every dimension
fits (halves index bandwidth of the HBM-bound SpMV), int64 otherwise.
"""
from __future__ import annotations

import numpy as np
import torch

# Value dtypes the compute kernels support.
VALUE_DTYPES = (torch.float32, torch.float64, torch.complex64, torch.complex128)
INDEX_DTYPES = (torch.int32, torch.int64)

coord_ty = np.int64
nnz_ty = np.uint64

_TORCH_FROM_NP = {
    np.dtype(np.float32): torch.float32,
    np.dtype(np.float64): torch.float64,
    np.dtype(np.complex64): torch.complex64,
    np.dtype(np.complex128): torch.complex128,
    np.dtype(np.int32): torch.int32,
    np.dtype(np.int64): torch.int64,
    np.dtype(np.int16): torch.int16,
    np.dtype(np.int8): torch.int8,
    np.dtype(np.uint8): torch.uint8,
    np.dtype(np.bool_): torch.bool,
    np.dtype(np.float16): torch.float16,
}
_NP_FROM_TORCH = {v: k for k, v in _TORCH_FROM_NP.items()}


def to_torch_dtype(dt) -> torch.dtype:
    if isinstance(dt, torch.dtype):
        return dt
    return _TORCH_FROM_NP[np.dtype(dt)]


def to_numpy_dtype(dt) -> np.dtype:
    if isinstance(dt, torch.dtype):
        return _NP_FROM_TORCH[dt]
    return np.dtype(dt)


def promote_value_dtype(dt: torch.dtype) -> torch.dtype:
    """Map an arbitrary input dtype onto one of the supported value dtypes."""
    if dt in VALUE_DTYPES:
        return dt
    if dt in (torch.int8, torch.int16, torch.int32, torch.int64, torch.uint8,
              torch.bool, torch.float16, torch.bfloat16):
        return torch.float64
    if dt == torch.complex32:
        return torch.complex64
    raise TypeError(f"unsupported value dtype {dt}")


def common_value_dtype(*dts: torch.dtype) -> torch.dtype:
    """scipy-style promotion across operand dtypes (sparse/utils.py:120-140)."""
    out = dts[0]
    for dt in dts[1:]:
        out = torch.promote_types(out, dt)
    return promote_value_dtype(out)


def index_dtype_for(shape) -> torch.dtype:
    return torch.int32 if max(shape, default=0) < 2**31 - 1 else torch.int64


def is_complex(dt: torch.dtype) -> bool:
    return dt in (torch.complex64, torch.complex128)
