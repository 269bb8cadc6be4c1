"""coo_array: COO triples partitioned by nnz position.

Reference parity: sparse/coo.py — holds (_i,_j,_vals) stores (coo.py:104-106);
tocsr/tocsc are the distributed sort path (coo.py:233-447: SORT_BY_KEY
samplesort + SORTED_COORDS_TO_COUNTS + nnz->pos).  Here the shuffle uses the
known row-owner map (alltoallv to owners + local torch sort), which is the
same communication volume without the sampling pass; parallel/sort.py holds
the general samplesort for unknown distributions.
"""
from __future__ import annotations

import numbers

import numpy as np
import torch

import scipy.sparse as _sps

from .coverage import clone_scipy_arr_kind
from .base import CompressedBase
from .darray import DistArray
from .ops import local as ops
from .parallel import comm
from .parallel.partition import RowPartition
from .parallel.shuffle import shuffle_to_owner
from .runtime import runtime
from .types import (
    index_dtype_for,
    promote_value_dtype,
    to_numpy_dtype,
    to_torch_dtype,
)


@clone_scipy_arr_kind(_sps.coo_matrix)
class coo_array(CompressedBase):
    _format = "coo"

    def __init__(self, arg, shape=None, dtype=None, copy=False):
        rt = runtime()
        tdtype = to_torch_dtype(dtype) if dtype is not None else None
        import scipy.sparse as sps

        if isinstance(arg, coo_array):
            v = arg._vals.clone() if copy else arg._vals
            self._init_from_local(arg._i, arg._j, v.to(tdtype) if tdtype else v, arg.shape)
            return
        if isinstance(arg, (sps.spmatrix, sps.sparray)):
            m = arg.tocoo()
            gshape = m.shape if shape is None else tuple(shape)
            self._init_replicated(m.row, m.col, m.data, gshape, tdtype)
            return
        if isinstance(arg, tuple) and len(arg) == 2 and isinstance(arg[1], tuple):
            data, (row, col) = arg
            data = np.asarray(data) if not isinstance(data, torch.Tensor) else data.cpu().numpy()
            row = np.asarray(row) if not isinstance(row, torch.Tensor) else row.cpu().numpy()
            col = np.asarray(col) if not isinstance(col, torch.Tensor) else col.cpu().numpy()
            if shape is None:
                m = int(row.max()) + 1 if row.size else 0
                n = int(col.max()) + 1 if col.size else 0
                shape = (m, n)
            self._init_replicated(row, col, data, tuple(shape), tdtype)
            return
        if isinstance(arg, tuple) and len(arg) == 2 and all(
                isinstance(a, numbers.Integral) for a in arg):
            gshape = tuple(int(a) for a in arg)
            idt = index_dtype_for(gshape)
            self._init_from_local(
                torch.zeros(0, dtype=idt, device=rt.device),
                torch.zeros(0, dtype=idt, device=rt.device),
                torch.zeros(0, dtype=tdtype or torch.float64, device=rt.device),
                gshape)
            return

        from .csc import csc_array
        from .csr import csr_array
        from .dia import dia_array

        if isinstance(arg, (csr_array, csc_array, dia_array)):
            c = arg.tocoo()
            v = c._vals.to(tdtype) if tdtype else c._vals
            self._init_from_local(c._i, c._j, v, c.shape)
            return
        # dense
        if isinstance(arg, DistArray):
            arg = arg.numpy()
        d = np.asarray(arg) if not isinstance(arg, torch.Tensor) else arg.cpu().numpy()
        if d.ndim != 2:
            raise ValueError("need a 2-D array to build a coo_array")
        m = sps.coo_matrix(d)
        self._init_replicated(m.row, m.col, m.data, tuple(d.shape) if shape is None else tuple(shape), tdtype)

    def _init_replicated(self, row, col, data, gshape, tdtype):
        """Every rank has the full triples; keep an equal nnz chunk."""
        rt = runtime()
        nnz = len(data)
        part = RowPartition.equal(nnz, comm.world_size())
        r = comm.rank()
        s, e = part.start(r), part.stop(r)
        idt = index_dtype_for(gshape)
        vals = torch.as_tensor(np.ascontiguousarray(data[s:e]), device=rt.device)
        vals = vals.to(tdtype) if tdtype else vals.to(promote_value_dtype(vals.dtype))
        self._init_from_local(
            torch.as_tensor(np.ascontiguousarray(row[s:e]), device=rt.device).to(idt),
            torch.as_tensor(np.ascontiguousarray(col[s:e]), device=rt.device).to(idt),
            vals, gshape)

    def _init_from_local(self, i, j, vals, shape):
        self._i = i
        self._j = j
        self._vals = vals
        self.shape = tuple(int(s) for s in shape)
        self._nnz_cache = None

    @classmethod
    def _from_local(cls, i, j, vals, shape) -> "coo_array":
        self = cls.__new__(cls)
        self._init_from_local(i, j, vals, shape)
        return self

    # -- properties -----------------------------------------------------------
    def _nnz_counts(self):
        t = torch.zeros(comm.world_size(), dtype=torch.int64)
        t[comm.rank()] = self._vals.numel()
        comm.all_reduce_(t)
        return [int(x) for x in t]

    @property
    def nnz(self) -> int:
        if self._nnz_cache is None:
            self._nnz_cache = sum(self._nnz_counts())
        return self._nnz_cache

    @property
    def dtype(self):
        return to_numpy_dtype(self._vals.dtype)

    @property
    def row(self) -> np.ndarray:
        return comm.all_gather_rows(self._i, self._nnz_counts()).cpu().numpy()

    @property
    def col(self) -> np.ndarray:
        return comm.all_gather_rows(self._j, self._nnz_counts()).cpu().numpy()

    @property
    def data(self) -> np.ndarray:
        return comm.all_gather_rows(self._vals, self._nnz_counts()).cpu().numpy()

    @data.setter
    def data(self, v):
        counts = self._nnz_counts()
        off = sum(counts[: comm.rank()])
        mine = np.asarray(v)[off: off + counts[comm.rank()]]
        self._vals = torch.as_tensor(mine, device=self._vals.device).to(self._vals.dtype)

    def _values_tensor(self):
        return self._vals

    def _with_values(self, fn) -> "coo_array":
        return coo_array._from_local(self._i, self._j, fn(self._vals), self.shape)

    # -- lifecycle ------------------------------------------------------------
    def copy(self) -> "coo_array":
        return coo_array._from_local(self._i.clone(), self._j.clone(),
                                     self._vals.clone(), self.shape)

    def astype(self, dtype, casting="unsafe", copy=True):
        t = to_torch_dtype(dtype)
        if t == self._vals.dtype and not copy:
            return self
        return self._with_values(lambda v: v.to(t))

    def conj(self, copy=True):
        if not self._vals.is_complex():
            return self.copy() if copy else self
        return self._with_values(lambda v: v.conj().resolve_conj())

    def power(self, n, dtype=None):
        t = to_torch_dtype(dtype) if dtype is not None else None
        return self._with_values(lambda v: (v.to(t) if t else v) ** n)

    def __neg__(self):
        return self._with_values(lambda v: -v)

    def transpose(self, copy=False):
        return coo_array._from_local(self._j, self._i, self._vals,
                                     (self.shape[1], self.shape[0]))

    @property
    def T(self):
        return self.transpose()

    def diagonal(self, k=0):
        """Reference: coo.py:180-197 (mask + scatter)."""
        dlen = min(self.shape[0] + min(k, 0), self.shape[1] - max(k, 0))
        if dlen <= 0:
            raise ValueError("k exceeds matrix dimensions")
        d = torch.zeros(dlen, dtype=self._vals.dtype, device=self._vals.device)
        hit = (self._j.long() - self._i.long()) == k
        rows = self._i[hit].long() + min(k, 0) * 0 - (0 if k >= 0 else -k) * 0
        # diag index: for k>=0 it's the row id; for k<0 it's the column id
        didx = self._i[hit].long() if k >= 0 else self._j[hit].long()
        d.index_add_(0, didx, self._vals[hit])
        comm.all_reduce_(d)
        return DistArray.from_global(d)

    # -- conversions (the distributed sort path) ------------------------------
    def tocsr(self, copy=False):
        from .csr import csr_array

        part = RowPartition.equal(self.shape[0], comm.world_size())
        i, j, v = shuffle_to_owner(self._i.to(torch.int64), part,
                                   self._j.to(torch.int64), self._vals)
        me = comm.rank()
        r0 = part.start(me)
        mloc = part.count(me)
        # segmented scatter+sort (GPU kernel) with dup-summing fallback
        idt = index_dtype_for(self.shape)
        indptr, cols, v = ops.local_coo_to_csr(
            i - r0, j.to(idt), v, mloc, self.shape[1])
        return csr_array.from_local(indptr, cols.to(idt), v, part, self.shape)

    def tocsc(self, copy=False):
        return self.transpose().tocsr().T

    def tocoo(self, copy=False):
        return self.copy() if copy else self

    def todia(self, copy=False):
        from .dia import dia_array

        # gather (small-matrix path, like the reference's DIA usage)
        import scipy.sparse as sps

        m = sps.coo_matrix((self.data, (self.row, self.col)), shape=self.shape).todia()
        return dia_array((m.data, m.offsets), shape=self.shape)

    def todense(self, order=None, out=None):
        """Reference: COO_TO_DENSE broadcast scatter (coo.py:449-465)."""
        d = torch.zeros(self.shape, dtype=self._vals.dtype, device=self._vals.device)
        flat = self._i.long() * self.shape[1] + self._j.long()
        d.view(-1).index_add_(0, flat, self._vals)
        comm.all_reduce_(d)
        res = DistArray.from_global(d)
        if out is not None:
            np.copyto(out, res.numpy())
            return out
        return res

    # -- products delegate to CSR (reference coo.py:467-477) ------------------
    def dot(self, other, out=None):
        return self.tocsr().dot(other, out=out)

    def __matmul__(self, other):
        return self.dot(other)

    def __rmatmul__(self, other):
        return self.tocsr().__rmatmul__(other)

    def matvec(self, other):
        return self.tocsr().matvec(other)

    def multiply(self, other):
        if isinstance(other, numbers.Number) or (
            isinstance(other, torch.Tensor) and other.dim() == 0):
            return self._with_values(lambda v: v * other)
        return self.tocsr().multiply(other)

    def __mul__(self, other):
        return self.multiply(other)

    __rmul__ = __mul__

    def __add__(self, other):
        if isinstance(other, numbers.Number):
            if other == 0:
                return self.copy()
            raise NotImplementedError("adding a nonzero scalar to a sparse matrix")
        return self.tocsr() + other

    __radd__ = __add__

    def __sub__(self, other):
        if isinstance(other, numbers.Number):
            if other == 0:
                return self.copy()
            raise NotImplementedError("subtracting a nonzero scalar")
        return self.tocsr() - other

    def __truediv__(self, other):
        if isinstance(other, numbers.Number) or (
            isinstance(other, torch.Tensor) and other.dim() == 0):
            return self._with_values(lambda v: v / other)
        raise NotImplementedError("sparse division by non-scalar")

    def __str__(self):
        import scipy.sparse as sps

        return str(sps.coo_matrix((self.data, (self.row, self.col)), shape=self.shape))


coo_matrix = coo_array
