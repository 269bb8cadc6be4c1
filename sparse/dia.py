"""dia_array: diagonal storage (data, offsets).

Reference parity: sparse/dia.py — tocoo masked expansion (dia.py:147-168),
fast transpose by flipping offsets + realigning data (dia.py:178-220),
tocsc via the vectorized scipy-style conversion (dia.py:222-249),
tocsr = T.tocsc().T (dia.py:175-176).

The data plane is replicated (offsets are few); DIA is a construction
format here — the compute paths go through CSR/CSC.  module.diags(format=
"csr") builds CSR slabs directly and never materializes a DIA.
"""
from __future__ import annotations

import numbers

import numpy as np
import torch

import scipy.sparse as _sps

from .coverage import clone_scipy_arr_kind
from .base import CompressedBase
from .runtime import runtime
from .types import promote_value_dtype, to_numpy_dtype, to_torch_dtype


@clone_scipy_arr_kind(_sps.dia_matrix)
class dia_array(CompressedBase):
    _format = "dia"

    def __init__(self, arg, shape=None, dtype=None, copy=False):
        rt = runtime()
        import scipy.sparse as sps

        if isinstance(arg, dia_array):
            self._data = arg._data.clone() if copy else arg._data
            self._offsets = arg._offsets.copy()
            self.shape = arg.shape
        elif isinstance(arg, (sps.spmatrix, sps.sparray)):
            m = arg.todia()
            self._data = torch.as_tensor(np.ascontiguousarray(m.data), device=rt.device)
            self._offsets = np.asarray(m.offsets, dtype=np.int64)
            self.shape = m.shape if shape is None else tuple(shape)
        elif isinstance(arg, tuple) and len(arg) == 2:
            data, offsets = arg
            data = np.atleast_2d(np.asarray(data) if not isinstance(data, torch.Tensor)
                                 else data.cpu().numpy())
            offsets = np.atleast_1d(np.asarray(offsets, dtype=np.int64))
            if shape is None:
                raise ValueError("dia_array((data, offsets)) requires shape")
            self._data = torch.as_tensor(np.ascontiguousarray(data), device=rt.device)
            self._offsets = offsets
            self.shape = tuple(shape)
        else:
            d = np.asarray(arg) if not isinstance(arg, torch.Tensor) else arg.cpu().numpy()
            m = sps.dia_matrix(d)
            self._data = torch.as_tensor(np.ascontiguousarray(m.data), device=rt.device)
            self._offsets = np.asarray(m.offsets, dtype=np.int64)
            self.shape = m.shape if shape is None else tuple(shape)
        if dtype is not None:
            self._data = self._data.to(to_torch_dtype(dtype))
        else:
            self._data = self._data.to(promote_value_dtype(self._data.dtype))
        self.shape = tuple(int(s) for s in self.shape)

    # -- properties -----------------------------------------------------------
    @property
    def data(self) -> np.ndarray:
        return self._data.cpu().numpy()

    @property
    def offsets(self) -> np.ndarray:
        return self._offsets

    @property
    def dtype(self):
        return to_numpy_dtype(self._data.dtype)

    @property
    def nnz(self) -> int:
        """Count of stored values inside the matrix bounds (scipy semantics,
        reference dia.py:118-128)."""
        m, n = self.shape
        total = 0
        for k in self._offsets:
            total += max(0, min(m + min(k, 0), n - max(k, 0)))
        return int(total)

    def _values_tensor(self):
        return self._data.reshape(-1)

    def _with_values(self, fn) -> "dia_array":
        out = dia_array((fn(self._data).cpu().numpy(), self._offsets), shape=self.shape)
        return out

    # -- lifecycle ------------------------------------------------------------
    def copy(self) -> "dia_array":
        return dia_array((self._data.clone().cpu().numpy(), self._offsets.copy()),
                         shape=self.shape)

    def astype(self, dtype, casting="unsafe", copy=True):
        t = to_torch_dtype(dtype)
        if t == self._data.dtype and not copy:
            return self
        return dia_array((self._data.to(t).cpu().numpy(), self._offsets), shape=self.shape)

    def conj(self, copy=True):
        if not self._data.is_complex():
            return self.copy() if copy else self
        return dia_array((self._data.conj().resolve_conj().cpu().numpy(), self._offsets),
                         shape=self.shape)

    def diagonal(self, k=0):
        from .darray import DistArray

        m, n = self.shape
        dlen = min(m + min(k, 0), n - max(k, 0))
        if dlen <= 0:
            raise ValueError("k exceeds matrix dimensions")
        out = torch.zeros(dlen, dtype=self._data.dtype, device=self._data.device)
        hits = np.where(self._offsets == k)[0]
        if hits.size:
            row = self._data[int(hits[0])]
            # dia data layout: data[d, j] is the value at column j of diag d;
            # the stored width may be SHORTER than s+dlen (scipy truncates
            # trailing zeros) — missing entries are zero
            s = max(k, 0)
            avail = max(0, min(dlen, row.shape[0] - s))
            out[:avail].copy_(row[s: s + avail])
        return DistArray.from_global(out)

    def transpose(self, axes=None, copy=False):
        """Flip offsets and realign each diagonal (reference dia.py:178-220)."""
        m, n = self.shape
        ndiag, width = self._data.shape
        new_width = max(m, n)
        data = self._data.cpu().numpy()
        new_data = np.zeros((ndiag, new_width), dtype=data.dtype)
        for d, k in enumerate(self._offsets):
            length = min(m + min(k, 0), n - max(k, 0))
            # stored width may truncate trailing zeros (scipy semantics)
            length = min(length, width - max(k, 0))
            if length <= 0:
                continue
            # entries of diag k live at data[d, max(k,0) : max(k,0)+length]
            src = data[d, max(k, 0): max(k, 0) + length]
            # in the transpose they are diag -k at cols max(-k,0)...
            new_data[d, max(-k, 0): max(-k, 0) + length] = src
        return dia_array((new_data, -self._offsets), shape=(n, m))

    @property
    def T(self):
        return self.transpose()

    # -- conversions ----------------------------------------------------------
    def tocoo(self, copy=False):
        """Masked expansion (reference dia.py:147-168)."""
        from .coo import coo_array

        m, n = self.shape
        data = self._data.cpu().numpy()
        rows_l, cols_l, vals_l = [], [], []
        for d, k in enumerate(self._offsets):
            length = min(m + min(k, 0), n - max(k, 0))
            # stored width may truncate trailing zeros (scipy semantics)
            length = min(length, data.shape[1] - max(k, 0))
            if length <= 0:
                continue
            j = np.arange(max(k, 0), max(k, 0) + length)
            i = j - k
            v = data[d, j]
            keep = v != 0
            rows_l.append(i[keep])
            cols_l.append(j[keep])
            vals_l.append(v[keep])
        if rows_l:
            rows = np.concatenate(rows_l)
            cols = np.concatenate(cols_l)
            vals = np.concatenate(vals_l)
        else:
            rows = np.zeros(0, dtype=np.int64)
            cols = np.zeros(0, dtype=np.int64)
            vals = np.zeros(0, dtype=self.dtype)
        return coo_array((vals, (rows, cols)), shape=self.shape, dtype=self.dtype)

    def tocsr(self):
        return self.tocoo().tocsr()

    def tocsc(self, copy=False):
        from .csc import csc_array

        return csc_array(self.tocoo())

    def todense(self, order=None, out=None):
        return self.tocoo().todense(order=order, out=out)

    def __mul__(self, other):
        if isinstance(other, numbers.Number):
            return dia_array(((self._data * other).cpu().numpy(), self._offsets),
                             shape=self.shape)
        return self.tocsr() * other

    __rmul__ = __mul__

    def __neg__(self):
        return dia_array(((-self._data).cpu().numpy(), self._offsets), shape=self.shape)

    def __add__(self, other):
        if isinstance(other, numbers.Number) and other == 0:
            return self.copy()
        return self.tocsr() + other

    def __sub__(self, other):
        if isinstance(other, numbers.Number) and other == 0:
            return self.copy()
        return self.tocsr() - other

    def dot(self, other, out=None):
        return self.tocsr().dot(other, out=out)

    def __matmul__(self, other):
        return self.dot(other)


dia_matrix = dia_array
