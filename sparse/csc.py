"""csc_array: column-partitioned CSC.

Reference parity: sparse/csc.py — the column-wise mirror of CSR.  A CSC
array IS the transpose view of a CSR array (reference csr.py:620-627 /
csc.py:317-340): we store local colptr over this rank's column slab, GLOBAL
row indices, values.  CSC SpMV computes a partial y over the local row
window and reduces into the owners (reference CSC_SPMV_COL_SPLIT,
src/sparse/array/csc/spmv.cu + csc.py:368-454), mapped here onto an explicit
ReduceScatterPlan over RCCL.
"""
from __future__ import annotations

import numbers

import numpy as np
import torch

import scipy.sparse as _sps

from .coverage import clone_scipy_arr_kind
from .base import CompressedBase, DenseSparseBase
from .darray import DistArray, asdistarray
from .ops import local as ops
from .parallel import comm
from .parallel.gather import ReduceScatterPlan, WindowGatherPlan
from .parallel.partition import RowPartition
from .runtime import runtime
from .types import (
    common_value_dtype,
    index_dtype_for,
    promote_value_dtype,
    to_numpy_dtype,
    to_torch_dtype,
)


@clone_scipy_arr_kind(_sps.csc_matrix)
class csc_array(CompressedBase, DenseSparseBase):
    _format = "csc"

    def __init__(self, arg, shape=None, dtype=None, copy=False):
        rt = runtime()
        tdtype = to_torch_dtype(dtype) if dtype is not None else None
        import scipy.sparse as sps

        if isinstance(arg, csc_array):
            v = arg._values.clone() if copy else arg._values
            self._init_from_local(arg._colptr, arg._indices,
                                  v.to(tdtype) if tdtype else v,
                                  arg.partition, arg.shape)
            return
        if isinstance(arg, (sps.spmatrix, sps.sparray)):
            m = arg.tocsc()
            m.sort_indices()
            gshape = m.shape if shape is None else tuple(shape)
            part = RowPartition.equal(gshape[1], comm.world_size())
            r = comm.rank()
            c0, c1 = part.start(r), part.stop(r)
            lptr = m.indptr[c0: c1 + 1].astype(np.int64)
            s, e = int(lptr[0]), int(lptr[-1])
            idt = index_dtype_for(gshape)
            vals = torch.as_tensor(m.data[s:e], device=rt.device)
            vals = vals.to(tdtype) if tdtype else vals.to(promote_value_dtype(vals.dtype))
            self._init_from_local(
                torch.as_tensor(lptr - s, dtype=torch.int64, device=rt.device),
                torch.as_tensor(m.indices[s:e], device=rt.device).to(idt),
                vals, part, gshape)
            return
        if isinstance(arg, tuple) and len(arg) == 3:
            data, indices, indptr = [np.asarray(a) if not isinstance(a, torch.Tensor)
                                     else a.cpu().numpy() for a in arg]
            n = indptr.shape[0] - 1
            mrows = shape[0] if shape is not None else (int(indices.max()) + 1 if indices.size else 0)
            gshape = (mrows, n) if shape is None else tuple(shape)
            sm = sps.csc_matrix((data, indices, indptr), shape=gshape)
            self.__init__(sm, dtype=dtype)
            return
        if isinstance(arg, tuple) and len(arg) == 2 and all(
                isinstance(a, numbers.Integral) for a in arg):
            gshape = tuple(int(a) for a in arg)
            part = RowPartition.equal(gshape[1], comm.world_size())
            nloc = part.count(comm.rank())
            self._init_from_local(
                torch.zeros(nloc + 1, dtype=torch.int64, device=rt.device),
                torch.zeros(0, dtype=index_dtype_for(gshape), device=rt.device),
                torch.zeros(0, dtype=tdtype or torch.float64, device=rt.device),
                part, gshape)
            return
        if isinstance(arg, tuple) and len(arg) == 2 and isinstance(arg[1], tuple):
            from .coo import coo_array

            c = coo_array(arg, shape=shape, dtype=dtype).tocsc()
            self._init_from_local(c._colptr, c._indices, c._values, c.partition, c.shape)
            return

        from .coo import coo_array
        from .csr import csr_array
        from .dia import dia_array

        if isinstance(arg, (coo_array, csr_array, dia_array)):
            c = arg.tocsc()
            v = c._values.to(tdtype) if tdtype else c._values
            self._init_from_local(c._colptr, c._indices, v, c.partition, c.shape)
            return
        # dense
        if isinstance(arg, DistArray):
            arg = arg.numpy()
        d = np.asarray(arg) if not isinstance(arg, torch.Tensor) else arg.cpu().numpy()
        if d.ndim != 2:
            raise ValueError("need a 2-D array to build a csc_array")
        self.__init__(sps.csc_matrix(d), shape=shape, dtype=dtype)

    def _init_from_local(self, colptr, indices, values, partition, shape):
        self._colptr = colptr
        self._indices = indices
        self._values = values
        self.partition = partition  # over COLUMNS
        self.shape = tuple(int(s) for s in shape)
        self._nnz_cache = None
        self._window_cache = None
        self._plan_cache = {}

    @classmethod
    def from_local(cls, colptr, indices, values, partition, shape) -> "csc_array":
        self = cls.__new__(cls)
        self._init_from_local(colptr, indices, values, partition, shape)
        return self

    # -- transpose views ------------------------------------------------------
    def transpose(self, copy=False):
        from .csr import csr_array

        if copy:
            return csr_array.from_local(self._colptr.clone(), self._indices.clone(),
                                        self._values.clone(), self.partition,
                                        (self.shape[1], self.shape[0]))
        return csr_array.from_local(self._colptr, self._indices, self._values,
                                    self.partition, (self.shape[1], self.shape[0]))

    @property
    def T(self):
        return self.transpose()

    # -- properties -----------------------------------------------------------
    @property
    def nnz(self) -> int:
        if self._nnz_cache is None:
            t = torch.tensor([self._values.numel()], dtype=torch.int64)
            comm.all_reduce_(t)
            self._nnz_cache = int(t.item())
        return self._nnz_cache

    @property
    def dtype(self):
        return to_numpy_dtype(self._values.dtype)

    def _nnz_counts(self):
        t = torch.zeros(comm.world_size(), dtype=torch.int64)
        t[comm.rank()] = self._values.numel()
        comm.all_reduce_(t)
        return [int(x) for x in t]

    @property
    def data(self) -> np.ndarray:
        return comm.all_gather_rows(self._values, self._nnz_counts()).cpu().numpy()

    @data.setter
    def data(self, v):
        counts = self._nnz_counts()
        off = sum(counts[: comm.rank()])
        mine = np.asarray(v)[off: off + counts[comm.rank()]]
        self._values = torch.as_tensor(mine, device=self._values.device).to(self._values.dtype)

    @property
    def indices(self) -> np.ndarray:
        return comm.all_gather_rows(self._indices, self._nnz_counts()).cpu().numpy()

    @property
    def indptr(self) -> np.ndarray:
        return self.T.indptr

    def _values_tensor(self):
        return self._values

    def _with_values(self, fn) -> "csc_array":
        return csc_array.from_local(self._colptr, self._indices, fn(self._values),
                                    self.partition, self.shape)

    def _local_row_nnz(self):
        return (self._colptr[1:] - self._colptr[:-1]).to(torch.int64)

    def _repartition(self, newpart):
        from .parallel.shuffle import repartition_csr

        cp, ix, vs = repartition_csr(self._colptr, self._indices, self._values,
                                     self.partition, newpart)
        self._init_from_local(cp, ix, vs, newpart, self.shape)

    # -- lifecycle ------------------------------------------------------------
    def copy(self) -> "csc_array":
        return csc_array.from_local(self._colptr.clone(), self._indices.clone(),
                                    self._values.clone(), self.partition, self.shape)

    def astype(self, dtype, casting="unsafe", copy=True):
        t = to_torch_dtype(dtype)
        if t == self._values.dtype and not copy:
            return self
        return self._with_values(lambda v: v.to(t))

    def conj(self, copy=True):
        if not self._values.is_complex():
            return self.copy() if copy else self
        return self._with_values(lambda v: v.conj().resolve_conj())

    def power(self, n, dtype=None):
        t = to_torch_dtype(dtype) if dtype is not None else None
        return self._with_values(lambda v: (v.to(t) if t else v) ** n)

    def __neg__(self):
        return self._with_values(lambda v: -v)

    # -- products (col-split with reduction, reference csc.py:368-454) --------
    def _row_window(self):
        if self._window_cache is None:
            if self._indices.numel() == 0:
                self._window_cache = (0, 0)
            else:
                self._window_cache = (int(self._indices.min().item()),
                                      int(self._indices.max().item()) + 1)
        return self._window_cache

    def dot(self, other, out=None):
        from .coo import coo_array
        from .csr import csr_array

        if isinstance(other, (csr_array, csc_array, coo_array)):
            return self.tocsr().dot(other)
        x = asdistarray(other)
        if x.ndim == 1:
            return self._spmv(x, out=out)
        if x.ndim == 2:
            return self._spmm(x, out=out)
        raise NotImplementedError

    def __matmul__(self, other):
        return self.dot(other)

    def __rmatmul__(self, other):
        A = asdistarray(other)
        if A.ndim == 1:
            return self.T.dot(A)
        return self.tocsr().__rmatmul__(other)

    def matvec(self, other, out=None):
        return self._spmv(asdistarray(other), out=out)

    def _gather_x_cols(self, x: DistArray):
        me = comm.rank()
        c0, c1 = self.partition.start(me), self.partition.stop(me)
        key = ("x", x.partition.starts)
        if key not in self._plan_cache:
            self._plan_cache[key] = WindowGatherPlan(c0, c1, x.partition)
        return self._plan_cache[key].gather(x.local)

    def _reduce_plan(self, ypart: RowPartition, rlo: int, rhi: int):
        key = ("y", ypart.starts, rlo, rhi)
        if key not in self._plan_cache:
            self._plan_cache[key] = ReduceScatterPlan(rlo, rhi, ypart)
        return self._plan_cache[key]

    def _spmv(self, x: DistArray, out=None) -> DistArray:
        if x.shape[0] != self.shape[1]:
            raise ValueError(f"dimension mismatch {self.shape} @ {x.shape}")
        xw = self._gather_x_cols(x)
        rlo, rhi = self._row_window()
        vdt = common_value_dtype(self._values.dtype, xw.dtype)
        partial = ops.csc_spmv(self._colptr, self._indices, self._values.to(vdt),
                               xw.to(vdt), rlo, rhi)
        ypart = RowPartition.equal(self.shape[0], comm.world_size())
        me = comm.rank()
        y = out if out is not None else DistArray.from_local(
            torch.zeros(ypart.count(me), dtype=vdt, device=self._values.device),
            ypart, (self.shape[0],))
        self._reduce_plan(y.partition, rlo, rhi).scatter_add(
            partial.to(y.local.dtype), y.local, beta=0.0 if out is not None else 1.0)
        return y

    def _spmm(self, B: DistArray, out=None) -> DistArray:
        if B.shape[0] != self.shape[1]:
            raise ValueError(f"dimension mismatch {self.shape} @ {B.shape}")
        Bw = self._gather_x_cols(B)
        rlo, rhi = self._row_window()
        vdt = common_value_dtype(self._values.dtype, Bw.dtype)
        partial = ops.csc_spmm(self._colptr, self._indices, self._values.to(vdt),
                               Bw.to(vdt), rlo, rhi)
        ypart = RowPartition.equal(self.shape[0], comm.world_size())
        me = comm.rank()
        y = out if out is not None else DistArray.from_local(
            torch.zeros((ypart.count(me), B.shape[1]), dtype=vdt,
                        device=self._values.device),
            ypart, (self.shape[0], B.shape[1]))
        self._reduce_plan(y.partition, rlo, rhi).scatter_add(
            partial.to(y.local.dtype), y.local, beta=0.0 if out is not None else 1.0)
        return y

    # -- conversions ----------------------------------------------------------
    def tocsc(self, copy=False):
        return self.copy() if copy else self

    def tocsr(self, copy=False):
        return self.T.tocsc().T

    def tocoo(self, copy=False):
        return self.T.tocoo().transpose()

    def todense(self, order=None, out=None):
        return self.tocsr().todense(order=order, out=out)

    def todia(self, copy=False):
        return self.tocsr().todia()

    def diagonal(self, k=0):
        if k != 0:
            return self.tocoo().diagonal(k=k)
        me = comm.rank()
        c0 = self.partition.start(me)
        d_local = ops.csr_diagonal(
            ops.LocalCSR(self._colptr, self._indices, self._values,
                         self.partition.count(me), self.shape[0]),
            row_offset=c0)
        dlen = min(self.shape)
        starts = [min(s, dlen) for s in self.partition.starts]
        part = RowPartition.from_starts(starts)
        return DistArray.from_local(d_local[: part.count(me)], part, (dlen,))

    def sddmm(self, C, D):
        """vals'[i,j] = vals[i,j] * (C[i,:] @ D[:,j]) on CSC structure
        (reference csc.py:495,556-...)."""
        from .parallel.gather import ColBlockGatherPlan

        C = asdistarray(C)
        D = asdistarray(D)
        me = comm.rank()
        c0, c1 = self.partition.start(me), self.partition.stop(me)
        # operand-block gathers (reference csr.py:1244-1312 mirrored):
        # D columns for MY col slab; C rows for my min/max ROW window only
        key = ("sddmm_d", D.partition.starts)
        if key not in self._plan_cache:
            self._plan_cache[key] = ColBlockGatherPlan(c0, c1, D.partition)
        Dblk = self._plan_cache[key].gather(D.local)  # (k, c1-c0)
        rlo, rhi = self._row_window()
        key = ("sddmm_c", C.partition.starts)
        if key not in self._plan_cache:
            self._plan_cache[key] = WindowGatherPlan(rlo, rhi, C.partition)
        Cblk = self._plan_cache[key].gather(C.local)  # (rhi-rlo, k)
        vdt = common_value_dtype(self._values.dtype,
                                 common_value_dtype(Cblk.dtype, Dblk.dtype))
        # local is CSR of A^T: rows = my cols j, entries at global rows i.
        # out[nz at (j,i)] = vals * (D^T[j,:] @ C^T[:,i - rlo])
        lc = ops.LocalCSR(self._colptr, self._indices, self._values.to(vdt),
                          c1 - c0, self.shape[0])
        out = ops.sddmm(lc, Dblk.T.contiguous().to(vdt),
                        Cblk.T.contiguous().to(vdt), col_lo=rlo)
        return csc_array.from_local(self._colptr, self._indices, out,
                                    self.partition, self.shape)

    # -- elementwise ----------------------------------------------------------
    def multiply(self, other):
        if isinstance(other, numbers.Number) or (
            isinstance(other, torch.Tensor) and other.dim() == 0):
            return self._with_values(lambda v: v * other)
        return self.tocsr().multiply(other).tocsc()

    def __mul__(self, other):
        return self.multiply(other)

    __rmul__ = __mul__

    def __add__(self, other):
        if isinstance(other, numbers.Number):
            if other == 0:
                return self.copy()
            raise NotImplementedError("adding a nonzero scalar to a sparse matrix")
        if isinstance(other, csc_array):
            return (self.T + other.T).T
        return self.tocsr() + other

    __radd__ = __add__

    def __sub__(self, other):
        if isinstance(other, numbers.Number):
            if other == 0:
                return self.copy()
            raise NotImplementedError("subtracting a nonzero scalar")
        if isinstance(other, csc_array):
            return (self.T - other.T).T
        return self.tocsr() - other

    def __truediv__(self, other):
        if isinstance(other, numbers.Number) or (
            isinstance(other, torch.Tensor) and other.dim() == 0):
            return self._with_values(lambda v: v / other)
        raise NotImplementedError("sparse division by non-scalar")

    def __str__(self):
        return str(self.tocsr().to_scipy_sparse_csr().tocsc())

    @classmethod
    def make_empty(cls, shape, dtype):
        return cls(tuple(shape), dtype=dtype)


csc_matrix = csc_array
