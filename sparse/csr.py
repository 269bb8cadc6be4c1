"""csr_array: row-partitioned CSR over one node's MI355X GPUs.

Reference parity: sparse/csr.py (the workhorse — constructors csr.py:100-262,
dot dispatch csr.py:442-582, conversions csr.py:587-686, elementwise
csr.py:724-775, free functions csr.py:863-1728).  Storage is scipy-style
(indptr/indices/values torch tensors per row slab, global column ids) —
SURVEY §7.1 — instead of Legion rect1-pos stores; distribution is an explicit
RowPartition + cached gather plans instead of Legion image partitions.
"""
from __future__ import annotations

import numbers
from typing import Optional

import os

import numpy as np
import torch

import scipy.sparse as _sps

from .coverage import clone_scipy_arr_kind
from .base import CompressedBase, DenseSparseBase
from .darray import DistArray, asdistarray
from .ops import local as ops
from .parallel import comm
from .parallel.gather import WindowGatherPlan
from .parallel.partition import RowPartition
from .parallel.shuffle import gather_csr_rows, repartition_csr, shuffle_to_owner
from .runtime import runtime
from .types import (
    common_value_dtype,
    index_dtype_for,
    promote_value_dtype,
    to_numpy_dtype,
    to_torch_dtype,
)


def _default_partition(m: int) -> RowPartition:
    return RowPartition.equal(m, comm.world_size())


@clone_scipy_arr_kind(_sps.csr_matrix)
class csr_array(CompressedBase, DenseSparseBase):
    _format = "csr"

    def __init__(self, arg, shape=None, dtype=None, copy=False):
        rt = runtime()
        tdtype = to_torch_dtype(dtype) if dtype is not None else None

        if isinstance(arg, csr_array):
            self._init_from_local(
                arg._indptr.clone() if copy else arg._indptr,
                (arg._indices.clone() if copy else arg._indices),
                (arg._values.clone() if copy else arg._values).to(tdtype) if tdtype else (
                    arg._values.clone() if copy else arg._values),
                arg.partition, arg.shape)
            return

        import scipy.sparse as sps

        if isinstance(arg, sps.spmatrix) or isinstance(arg, sps.sparray):
            m = arg.tocsr()
            m.sort_indices()
            gshape = m.shape if shape is None else shape
            part = _default_partition(gshape[0])
            r = comm.rank()
            r0, r1 = part.start(r), part.stop(r)
            lptr = m.indptr[r0: r1 + 1].astype(np.int64)
            s, e = int(lptr[0]), int(lptr[-1])
            idt = index_dtype_for(gshape)
            vals = torch.as_tensor(m.data[s:e], device=rt.device)
            if tdtype:
                vals = vals.to(tdtype)
            else:
                vals = vals.to(promote_value_dtype(vals.dtype))
            self._init_from_local(
                torch.as_tensor(lptr - s, dtype=torch.int64, device=rt.device),
                torch.as_tensor(m.indices[s:e], device=rt.device).to(idt),
                vals, part, tuple(gshape))
            return

        # (data, indices, indptr) or (data, (row, col)) or shape tuple
        if isinstance(arg, tuple):
            if len(arg) == 2 and all(isinstance(a, numbers.Integral) for a in arg):
                # empty matrix of given shape
                gshape = tuple(int(a) for a in arg)
                part = _default_partition(gshape[0])
                mloc = part.count(comm.rank())
                idt = index_dtype_for(gshape)
                vdt = tdtype or torch.float64
                self._init_from_local(
                    torch.zeros(mloc + 1, dtype=torch.int64, device=rt.device),
                    torch.zeros(0, dtype=idt, device=rt.device),
                    torch.zeros(0, dtype=vdt, device=rt.device),
                    part, gshape)
                return
            if len(arg) == 2 and isinstance(arg[1], tuple):
                # COO-style (data, (row, col))
                from .coo import coo_array

                c = coo_array(arg, shape=shape, dtype=dtype)
                r = c.tocsr()
                self._init_from_local(r._indptr, r._indices, r._values, r.partition, r.shape)
                return
            if len(arg) == 3:
                data, indices, indptr = arg
                if isinstance(indptr, torch.Tensor) and shape is not None and (
                    indptr.numel() == _default_partition(shape[0]).count(comm.rank()) + 1
                    and indptr.numel() != shape[0] + 1
                ):
                    # already-local torch tensors (internal fast path)
                    part = _default_partition(shape[0])
                    self._init_from_local(indptr.to(torch.int64), indices, data, part, shape)
                    return
                data = np.asarray(data) if not isinstance(data, torch.Tensor) else data.cpu().numpy()
                indices = np.asarray(indices) if not isinstance(indices, torch.Tensor) else indices.cpu().numpy()
                indptr = np.asarray(indptr) if not isinstance(indptr, torch.Tensor) else indptr.cpu().numpy()
                m = indptr.shape[0] - 1
                n = shape[1] if shape is not None else (int(indices.max()) + 1 if indices.size else 0)
                gshape = (m, n) if shape is None else tuple(shape)
                part = _default_partition(gshape[0])
                r = comm.rank()
                r0, r1 = part.start(r), part.stop(r)
                lptr = indptr[r0: r1 + 1].astype(np.int64)
                s, e = int(lptr[0]), int(lptr[-1])
                idt = index_dtype_for(gshape)
                vals = torch.as_tensor(np.ascontiguousarray(data[s:e]), device=rt.device)
                vals = vals.to(tdtype) if tdtype else vals.to(promote_value_dtype(vals.dtype))
                self._init_from_local(
                    torch.as_tensor(lptr - s, dtype=torch.int64, device=rt.device),
                    torch.as_tensor(np.ascontiguousarray(indices[s:e]), device=rt.device).to(idt),
                    vals, part, gshape)
                return
            raise NotImplementedError(f"cannot construct csr_array from tuple of len {len(arg)}")

        from .coo import coo_array
        from .csc import csc_array
        from .dia import dia_array

        if isinstance(arg, (coo_array, csc_array, dia_array)):
            r = arg.tocsr()
            vals = r._values.to(tdtype) if tdtype else r._values
            self._init_from_local(r._indptr, r._indices, vals, r.partition, r.shape)
            return

        # dense (numpy 2-D / torch 2-D / DistArray 2-D)
        if isinstance(arg, DistArray):
            if arg.ndim != 2:
                raise ValueError("need a 2-D array to build a csr_array")
            part = arg.partition
            lc = ops.dense_to_csr(arg.local, ncols=arg.shape[1])
            vals = lc.values.to(tdtype) if tdtype else lc.values.to(promote_value_dtype(lc.values.dtype))
            self._init_from_local(lc.indptr, lc.indices, vals, part, arg.shape)
            return
        d = torch.as_tensor(np.asarray(arg)) if not isinstance(arg, torch.Tensor) else arg
        if d.dim() != 2:
            raise ValueError("need a 2-D array to build a csr_array")
        gshape = tuple(d.shape)
        part = _default_partition(gshape[0])
        r = comm.rank()
        dloc = d[part.start(r): part.stop(r)].to(rt.device)
        lc = ops.dense_to_csr(dloc, ncols=gshape[1])
        vals = lc.values.to(tdtype) if tdtype else lc.values.to(promote_value_dtype(lc.values.dtype))
        self._init_from_local(lc.indptr, lc.indices, vals, part, gshape)

    # -- internal -------------------------------------------------------------
    def _init_from_local(self, indptr, indices, values, partition, shape):
        import os as _os

        if _os.environ.get("SPARSE_BOUNDS_CHECKS"):
            # debug-mode accessor checks (reference Legion_BOUNDS_CHECKS,
            # legate_sparse_cpp.cmake:62)
            assert indptr.numel() == partition.count(comm.rank()) + 1, \
                "indptr length != local rows + 1"
            assert int(indptr[0].item()) == 0, "indptr[0] != 0"
            assert bool((indptr[1:] >= indptr[:-1]).all().item()), \
                "indptr not monotone"
            assert int(indptr[-1].item()) == indices.numel() == values.numel(), \
                "nnz mismatch between indptr/indices/values"
            if indices.numel():
                assert int(indices.min().item()) >= 0 and \
                    int(indices.max().item()) < shape[1], "column out of bounds"
        self._indptr = indptr
        self._indices = indices
        self._values = values
        self.partition = partition
        self.shape = tuple(int(s) for s in shape)
        self._nnz_cache = None
        self._window_cache = None
        self._plan_cache = {}
        self._ell_cache = None
        self._dia_cache = None
        self._bsr_cache = None
        self._csc_cache = None
        self._maxrow_cache = None

    @classmethod
    def from_local(cls, indptr, indices, values, partition, shape) -> "csr_array":
        self = cls.__new__(cls)
        CompressedBase.__init__(self)
        self._init_from_local(indptr, indices, values, partition, shape)
        return self

    @property
    def local(self) -> ops.LocalCSR:
        return ops.LocalCSR(self._indptr, self._indices, self._values,
                            self.partition.count(comm.rank()), self.shape[1],
                            self._maxrow_cache)

    def _max_row_nnz(self) -> int:
        if self._maxrow_cache is None:
            d = self._indptr[1:] - self._indptr[:-1]
            self._maxrow_cache = int(d.max().item()) if d.numel() else 0
        return self._maxrow_cache

    def _col_window(self):
        """[lo,hi) min/max column window of the local slab (the MinMaxImage
        bound, partition.py:139-208)."""
        if self._window_cache is None:
            if self._indices.numel() == 0:
                self._window_cache = (0, 0)
            else:
                self._window_cache = (int(self._indices.min().item()),
                                      int(self._indices.max().item()) + 1)
        return self._window_cache

    def __getitem__(self, key):
        """Minimal indexing (scipy-API superset; the reference supports no
        indexing): A[i] -> 1-row csr, A[i:j] -> row-slice csr (step 1),
        A[i, j] -> scalar.  Row slices repartition collectively."""
        from .parallel.shuffle import gather_csr_rows

        if isinstance(key, tuple) and len(key) == 2 and all(
                isinstance(k, (int, np.integer)) for k in key):
            i, j = int(key[0]), int(key[1])
            if i < 0:
                i += self.shape[0]
            if j < 0:
                j += self.shape[1]
            row = self[i]
            idx = row._indices
            vals = row._values
            hit = (idx.long() == j).nonzero(as_tuple=True)[0]
            v = vals[hit[0]].item() if hit.numel() else self.dtype.type(0)
            # replicated row: every rank returns the same scalar
            return self.dtype.type(v)
        if isinstance(key, (int, np.integer)):
            i = int(key)
            if i < 0:
                i += self.shape[0]
            if not 0 <= i < self.shape[0]:
                raise IndexError(f"row {key} out of range")
            key = slice(i, i + 1)
        if isinstance(key, slice):
            start, stop, step = key.indices(self.shape[0])
            if step != 1:
                raise NotImplementedError("only step-1 row slices")
            nrows = max(0, stop - start)
            part = RowPartition.equal(nrows, comm.world_size())
            me = comm.rank()
            lo = start + part.start(me)
            hi = start + part.stop(me)
            ip, ix, vs = gather_csr_rows(self._indptr, self._indices,
                                         self._values, self.partition, lo, hi)
            return csr_array.from_local(ip, ix, vs, part,
                                        (nrows, self.shape[1]))
        raise NotImplementedError(f"indexing with {key!r} is not supported")

    def _eliminate_zeros_impl(self, keep):
        lc = self.local
        counts = lc.indptr[1:] - lc.indptr[:-1]
        rows = torch.repeat_interleave(
            torch.arange(lc.nrows, dtype=torch.int64, device=keep.device),
            counts)[keep]
        self._indices = self._indices[keep]
        self._values = self._values[keep]
        newip = torch.zeros(lc.nrows + 1, dtype=torch.int64,
                            device=self._indptr.device)
        if rows.numel():
            torch.cumsum(torch.bincount(rows, minlength=lc.nrows), 0,
                         out=newip[1:])
        self._indptr = newip
        self._invalidate_caches()

    def _invalidate_caches(self):
        self._ell_cache = None
        self._dia_cache = None
        self._bsr_cache = None
        self._csc_cache = None
        self._maxrow_cache = None
        self._nnz_cache = None
        self._window_cache = None
        self._plan_cache = {}

    def _dia_interior(self, dm, plan):
        """Even row bounds [a, b) of the interior — rows whose whole
        window falls inside the own piece, safe to compute while the halo
        exchange is in flight (overlap of comm with interior compute)."""
        me = comm.rank()
        xs, xe = plan.xpart.start(me), plan.xpart.stop(me)
        own_a, own_b = max(plan.lo, xs), min(plan.hi, xe)
        row0 = self.partition.start(me)
        a = min(max(0, own_a - row0 - dm.off_min), dm.m)
        b = min(max(a, own_b - row0 - dm.off_max), dm.m)
        a = (a + 1) // 2 * 2
        b = max(a, b // 2 * 2)
        return a, b

    def _ell_interior(self, ell, plan):
        """Even row bounds [a, b) of the longest contiguous interior run —
        rows whose whole column window lies in the own x piece (the ELL
        analog of _dia_interior; (0, 0) disables the overlap split)."""
        me = comm.rank()
        key = ("ell_int", plan.lo, plan.hi, plan.xpart.starts)
        if key not in self._plan_cache:
            from . import kernels

            xs, xe = plan.xpart.start(me), plan.xpart.stop(me)
            own_a = max(plan.lo, xs)
            own_b = max(own_a, min(plan.hi, xe))
            self._plan_cache[key] = kernels.ell_interior(ell, own_a, own_b)
        return self._plan_cache[key]

    def _dia(self):
        """Cached diagonal mirror (GPU banded fast SpMV; kernels.build_dia —
        values only, no index stream)."""
        if not self._values.is_cuda or self._dia_cache == "no":
            return None
        if self._dia_cache is None:
            from . import kernels

            kernels.require()
            dm = kernels.build_dia(self.local, self.partition.start(comm.rank()))
            self._dia_cache = dm or "no"
        return None if self._dia_cache == "no" else self._dia_cache

    def _bsr(self):
        """Cached 16x16-block MFMA mirror for multi-vector SpMM
        (kernels.build_bsr; measured win region in profiles/MFMA_r02.md)."""
        if not self._values.is_cuda or self._bsr_cache == "no":
            return None
        if self._bsr_cache is None:
            from . import kernels

            kernels.require()
            bm = kernels.build_bsr(self.local)
            self._bsr_cache = bm or "no"
        return None if self._bsr_cache == "no" else self._bsr_cache

    def _ell(self):
        """Cached padded-ELL mirror (GPU fast SpMV; kernels.build_ell)."""
        if not self._values.is_cuda or self._ell_cache == "no":
            return None
        if self._ell_cache is None:
            from . import kernels

            kernels.require()
            self._ell_cache = kernels.build_ell(self.local) or "no"
        return None if self._ell_cache == "no" else self._ell_cache

    def _xplan(self, xpart: RowPartition) -> WindowGatherPlan:
        from .settings import settings

        precise = settings.precise_images and comm.world_size() > 1
        key = ("x", xpart.starts, precise)
        if key not in self._plan_cache:
            if precise:
                from .parallel.gather import PreciseGatherPlan

                self._plan_cache[key] = PreciseGatherPlan(self._indices, xpart)
            else:
                lo, hi = self._col_window()
                self._plan_cache[key] = WindowGatherPlan(lo, hi, xpart)
        return self._plan_cache[key]

    # -- basic properties ----------------------------------------------------
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

    _PROP_WARN_NNZ = 50_000_000

    def _warn_global_read(self, what: str) -> None:
        # O(global) host materialization: cheap correctness for tests and
        # small matrices, a trap at scale — warn once
        # per matrix per property past a size threshold
        if self.nnz >= self._PROP_WARN_NNZ and what not in getattr(
                self, "_prop_warned", set()):
            from .utils import perf_warning

            if not hasattr(self, "_prop_warned"):
                self._prop_warned = set()
            self._prop_warned.add(what)
            perf_warning(
                f".{what} gathers the ENTIRE distributed array to host "
                f"({self.nnz} nnz); use .local / to_scipy_sparse_csr() on "
                f"a subset, or keep computation in sparse ops")

    @property
    def data(self) -> np.ndarray:
        self._warn_global_read("data")
        return comm.all_gather_rows(self._values, self._nnz_counts()).cpu().numpy()

    @data.setter
    def data(self, v):
        counts = self._nnz_counts()
        off = sum(counts[: comm.rank()])
        mine = np.asarray(v)[off: off + counts[comm.rank()]]
        self._values = torch.as_tensor(mine, device=self._values.device).to(self._values.dtype)
        self._ell_cache = None
        self._dia_cache = None
        self._bsr_cache = None
        self._csc_cache = None

    @property
    def indices(self) -> np.ndarray:
        self._warn_global_read("indices")
        return comm.all_gather_rows(self._indices, self._nnz_counts()).cpu().numpy()

    @property
    def indptr(self) -> np.ndarray:
        # reassemble the global indptr from per-rank local ones
        if comm.world_size() == 1:
            return self._indptr.cpu().numpy()
        self._warn_global_read("indptr")
        counts = self._nnz_counts()
        local = (self._indptr[1:]).cpu()
        glob = comm.all_gather_rows(local, self.partition.counts()).numpy().astype(np.int64)
        offs = np.zeros(len(glob) + 1, dtype=np.int64)
        # per-rank local indptrs restart at 0; add rank nnz offsets
        ws = comm.world_size()
        out = np.zeros(self.shape[0] + 1, dtype=np.int64)
        pos = 0
        base = 0
        for rk in range(ws):
            c = self.partition.count(rk)
            seg = glob[pos: pos + c]
            out[pos + 1: pos + c + 1] = seg + base
            base += counts[rk]
            pos += c
        return out

    def _nnz_counts(self):
        t = torch.zeros(comm.world_size(), dtype=torch.int64)
        t[comm.rank()] = self._values.numel()
        comm.all_reduce_(t)
        return [int(x) for x in t]

    def _values_tensor(self):
        return self._values

    def _with_values(self, fn) -> "csr_array":
        return csr_array.from_local(self._indptr, self._indices, fn(self._values),
                                    self.partition, self.shape)

    def _local_row_nnz(self):
        return (self._indptr[1:] - self._indptr[:-1]).to(torch.int64)

    def _repartition(self, newpart: RowPartition):
        ip, ix, vs = repartition_csr(self._indptr, self._indices, self._values,
                                     self.partition, newpart)
        self._init_from_local(ip, ix, vs, newpart, self.shape)

    # -- lifecycle ------------------------------------------------------------
    def copy(self) -> "csr_array":
        return csr_array.from_local(self._indptr.clone(), self._indices.clone(),
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

    def __pos__(self):
        return self.copy()

    def to_scipy_sparse_csr(self):
        import scipy.sparse as sps

        if comm.world_size() == 1:
            lc = self.local
            return lc.to_scipy()
        ip = self.indptr
        return sps.csr_matrix((self.data, self.indices, ip), shape=self.shape)

    # -- products -------------------------------------------------------------
    def dot(self, other, out=None, spmv_domain_part=False):
        from .coo import coo_array
        from .csc import csc_array

        if isinstance(other, (csr_array, csc_array, coo_array)):
            if isinstance(other, csc_array) and comm.world_size() > 1:
                return self._spgemm_2d(other)
            B = other.tocsr() if not isinstance(other, csr_array) else other
            return self._spgemm(B)
        x = other
        if isinstance(x, (np.ndarray, list)):
            x = asdistarray(np.asarray(x))
        elif isinstance(x, torch.Tensor):
            x = asdistarray(x)
        if isinstance(x, DistArray):
            if x.ndim == 1:
                if spmv_domain_part and comm.world_size() > 1:
                    # domain (column) partition SpMV: each rank multiplies
                    # its COLUMN block against its own x slab (no x gather)
                    # and partial-y windows are sum-reduced to the row
                    # owners — the reference's CSR_SPMV_COL_SPLIT strategy
                    # (csr.py:869-927), executed via a cached CSC mirror +
                    # ReduceScatterPlan.
                    if self._csc_cache is None:
                        self._csc_cache = self.tocsc()
                    return self._csc_cache._spmv(x, out=out)
                return self._spmv(x, out=out)
            if x.ndim == 2:
                return self._spmm(x, out=out)
        raise NotImplementedError(f"csr_array.dot with {type(other)}")

    def matvec(self, other, out=None):
        return self._spmv(asdistarray(other), out=out)

    def __matmul__(self, other):
        return self.dot(other)

    def __rmatmul__(self, other):
        # dense @ csr -> rspmm (reference csr.py:778-797, 1209-1240)
        A = asdistarray(other)
        if A.ndim == 1:
            # x @ A == (A.T x)
            return self.T.dot(A)
        return self._rspmm(A)

    def _out_dtype(self, other_dtype) -> torch.dtype:
        return common_value_dtype(self._values.dtype, other_dtype)

    def _spmv(self, x: DistArray, out: Optional[DistArray] = None,
              y_part: Optional[RowPartition] = None) -> DistArray:
        if x.shape[0] != self.shape[1]:
            raise ValueError(f"dimension mismatch {self.shape} @ {x.shape}")
        plan = self._xplan(x.partition)
        precise = not isinstance(plan, WindowGatherPlan)
        vdt = self._out_dtype(x.local.dtype)
        dm = (self._dia() if self._values.dtype == vdt and not precise
              else None)
        ell = None if dm is not None else (
            self._ell() if self._values.dtype == vdt and not precise
            else None)
        # write straight into out when layouts agree — saves a full
        # vector pass per matvec(out=) (the CG/GMG hot loops)
        mloc = self.partition.count(comm.rank())

        def _ybuf():
            if (out is not None and out.local.dtype == vdt
                    and out.local.is_contiguous()
                    and out.local.numel() == mloc):
                return out.local
            return torch.empty(mloc, dtype=vdt, device=self._values.device)

        if dm is not None:
            from . import kernels

            ylocal = _ybuf()
            ws_ = plan.hi - plan.lo
            if comm.world_size() > 1 and not os.environ.get("SPARSE_NO_OVERLAP"):
                h = plan.gather_halos_begin(x.local.to(vdt))
                a, bnd = self._dia_interior(dm, plan)
                kernels.dia_spmv(dm, plan.handle_pieces(h), ylocal, plan.lo,
                                 ws_, a, bnd)
                pieces = plan.gather_halos_end(h)
                kernels.dia_spmv(dm, pieces, ylocal, plan.lo, ws_, 0, a)
                kernels.dia_spmv(dm, pieces, ylocal, plan.lo, ws_, bnd, -1)
            else:
                pieces = plan.gather_halos(x.local.to(vdt))
                kernels.dia_spmv(dm, pieces, ylocal, plan.lo, ws_)
        elif ell is not None:
            from . import kernels

            ylocal = _ybuf()
            a = b = 0
            if comm.world_size() > 1 and not os.environ.get("SPARSE_NO_OVERLAP"):
                a, b = self._ell_interior(ell, plan)
            if b > a:
                h = plan.gather_halos_begin(x.local.to(vdt))
                kernels.ell_spmv(ell, plan.handle_pieces(h), ylocal,
                                 plan.lo, a, b)
                pieces = plan.gather_halos_end(h)
                kernels.ell_spmv(ell, pieces, ylocal, plan.lo, 0, a)
                kernels.ell_spmv(ell, pieces, ylocal, plan.lo, b, -1)
            else:
                pieces = plan.gather_halos(x.local.to(vdt))
                kernels.ell_spmv(ell, pieces, ylocal, plan.lo)
        else:
            xw = plan.gather(x.local.to(vdt))
            self._max_row_nnz()
            lc = self.local
            if precise:
                lc = ops.LocalCSR(lc.indptr, plan.remap(lc.indices),
                                  lc.values, lc.nrows, plan.hi,
                                  lc.max_row_nnz)
            if lc.values.dtype != vdt:
                lc = ops.LocalCSR(lc.indptr, lc.indices, lc.values.to(vdt),
                                  lc.nrows, lc.ncols, lc.max_row_nnz)
            yb = _ybuf()
            if yb.is_cuda:
                ylocal = ops.spmv(lc, xw, col_lo=plan.lo, y=yb)
            else:
                ylocal = ops.spmv(lc, xw, col_lo=plan.lo)
        part = self.partition
        if out is not None:
            if ylocal.data_ptr() != out.local.data_ptr():
                out.local.copy_(ylocal.to(out.local.dtype))
            return out
        return DistArray.from_local(ylocal, part, (self.shape[0],))

    def _spmm(self, B: DistArray, out: Optional[DistArray] = None) -> DistArray:
        if B.shape[0] != self.shape[1]:
            raise ValueError(f"dimension mismatch {self.shape} @ {B.shape}")
        plan = self._xplan(B.partition)
        vdt = self._out_dtype(B.local.dtype)
        k = int(B.local.shape[1]) if B.local.dim() == 2 else 1
        if (self._values.is_cuda and isinstance(plan, WindowGatherPlan)
                and k >= 16 and vdt == self._values.dtype
                and vdt in (torch.float64, torch.float32)
                and not os.environ.get("SPARSE_NO_BSR")):
            from . import kernels

            bm = self._bsr()
            use_bsr = kernels.bsr_profitable(bm, k)
            if comm.world_size() > 1:
                # block fill differs per slab: the aligned-window plan
                # construction below is COLLECTIVE, so the route must be
                # unanimous (all-reduce(min) vote, cached — profitability
                # depends only on the fixed fill and the k<=32 bucket)
                vkey = ("bsr_vote", k <= 32)
                if vkey not in self._plan_cache:
                    flag = torch.tensor([1.0 if use_bsr else 0.0],
                                        device=self._values.device)
                    comm.all_reduce_(flag, op="min")
                    self._plan_cache[vkey] = bool(flag.item() > 0.5)
                use_bsr = self._plan_cache[vkey]
            if use_bsr:
                # MFMA path reads whole 16-row blocks of B: use a
                # block-aligned window (profiles/MFMA_r02.md win region)
                lo, hi = self._col_window()
                lo16 = (lo >> 4) << 4
                hi16 = min(self.shape[1], ((hi + 15) >> 4) << 4)
                key = ("x16", B.partition.starts)
                if key not in self._plan_cache:
                    self._plan_cache[key] = WindowGatherPlan(
                        lo16, hi16, B.partition)
                aplan = self._plan_cache[key]
                Bw = aplan.gather(B.local.to(vdt)).contiguous()
                mloc = self.partition.count(comm.rank())
                if (out is not None and out.local.dtype == vdt
                        and out.local.is_contiguous()
                        and tuple(out.local.shape) == (mloc, k)):
                    Clocal = out.local  # direct write: no output copy pass
                else:
                    Clocal = torch.empty((mloc, k), dtype=vdt,
                                         device=self._values.device)
                kernels.bsr_spmm(bm, Bw, Clocal, aplan.lo)
                if out is not None:
                    if Clocal.data_ptr() != out.local.data_ptr():
                        out.local.copy_(Clocal.to(out.local.dtype))
                    return out
                return DistArray.from_local(Clocal, self.partition,
                                            (self.shape[0], k))
        Bw = plan.gather(B.local.to(vdt))
        lc = self.local
        if not isinstance(plan, WindowGatherPlan):
            lc = ops.LocalCSR(lc.indptr, plan.remap(lc.indices), lc.values,
                              lc.nrows, plan.hi, lc.max_row_nnz)
        if lc.values.dtype != vdt:
            lc = ops.LocalCSR(lc.indptr, lc.indices, lc.values.to(vdt),
                              lc.nrows, lc.ncols, lc.max_row_nnz)
        Cbuf = None
        if (out is not None and out.local.dtype == vdt
                and out.local.is_contiguous() and out.local.dim() == 2
                and out.local.is_cuda):
            Cbuf = out.local
        Clocal = ops.spmm(lc, Bw, col_lo=plan.lo, C=Cbuf)
        if out is not None:
            if Clocal.data_ptr() != out.local.data_ptr():
                out.local.copy_(Clocal.to(out.local.dtype))
            return out
        return DistArray.from_local(Clocal, self.partition, (self.shape[0], B.shape[1]))

    def jacobi_smooth(self, x: DistArray, b: DistArray, dinv: DistArray,
                      omega: float, out: Optional[DistArray] = None) -> DistArray:
        """One fused weighted-Jacobi sweep x' = x + omega*dinv*(b - A x)
        (the GMG/AMG smoother; reference WeightedJacobi gmg.py:247-285).
        Single kernel on the ELL fast path; generic fallback otherwise."""
        dm = self._dia()
        ell = None if dm is not None else self._ell()
        if out is None:
            out = DistArray.from_local(torch.empty_like(x.local), x.partition,
                                       x.gshape)
        if dm is not None:
            from . import kernels

            plan = self._xplan(x.partition)
            ws_ = plan.hi - plan.lo
            if comm.world_size() > 1 and not os.environ.get("SPARSE_NO_OVERLAP"):
                h = plan.gather_halos_begin(x.local)
                a, bnd = self._dia_interior(dm, plan)
                kernels.dia_jacobi(dm, plan.handle_pieces(h), x.local,
                                   b.local, dinv.local, omega, out.local,
                                   plan.lo, ws_, a, bnd)
                pieces = plan.gather_halos_end(h)
                kernels.dia_jacobi(dm, pieces, x.local, b.local, dinv.local,
                                   omega, out.local, plan.lo, ws_, 0, a)
                kernels.dia_jacobi(dm, pieces, x.local, b.local, dinv.local,
                                   omega, out.local, plan.lo, ws_, bnd, -1)
            else:
                pieces = plan.gather_halos(x.local)
                kernels.dia_jacobi(dm, pieces, x.local, b.local, dinv.local,
                                   omega, out.local, plan.lo, ws_)
            return out
        if ell is not None:
            from . import kernels

            plan = self._xplan(x.partition)
            pieces = plan.gather_halos(x.local)
            kernels.ell_jacobi(ell, pieces, x.local, b.local, dinv.local,
                               omega, out.local, plan.lo)
            return out
        r = self._spmv(x)
        r.local.sub_(b.local).neg_()  # r = b - A x
        out.local.copy_(x.local)
        out.local.addcmul_(r.local, dinv.local, value=omega)
        return out

    def residual(self, x: DistArray, b: DistArray,
                 out: Optional[DistArray] = None) -> DistArray:
        """r = b - A @ x in ONE kernel on the DIA fast path (the V-cycle
        residual; saves the separate 3-pass pointwise subtract of the
        naive r = A.dot(x); r -= b; r.neg_()).  Falls back to SpMV +
        pointwise elsewhere.  Reference context: the implicit copy+ADD
        chain of the reference's GMG residual (examples/gmg.py:118-119
        there)."""
        dm = self._dia()
        if out is None:
            out = DistArray.from_local(
                torch.empty(self.partition.count(comm.rank()),
                            dtype=self._values.dtype,
                            device=self._values.device),
                self.partition, (self.shape[0],))
        if dm is not None:
            from . import kernels

            plan = self._xplan(x.partition)
            ws_ = plan.hi - plan.lo
            if comm.world_size() > 1 and not os.environ.get("SPARSE_NO_OVERLAP"):
                h = plan.gather_halos_begin(x.local)
                a, bnd = self._dia_interior(dm, plan)
                kernels.dia_residual(dm, plan.handle_pieces(h), b.local,
                                     out.local, plan.lo, ws_, a, bnd)
                pieces = plan.gather_halos_end(h)
                kernels.dia_residual(dm, pieces, b.local, out.local,
                                     plan.lo, ws_, 0, a)
                kernels.dia_residual(dm, pieces, b.local, out.local,
                                     plan.lo, ws_, bnd, -1)
            else:
                pieces = plan.gather_halos(x.local)
                kernels.dia_residual(dm, pieces, b.local, out.local,
                                     plan.lo, ws_)
            return out
        self._spmv(x, out=out)
        out.local.sub_(b.local).neg_()
        return out

    def spmv_dot(self, p: DistArray, q: DistArray) -> torch.Tensor:
        """Fused q = A@p and all-reduced sum(p*q) — the CG p·Ap in one kernel
        (GPU, real dtypes; MI355X fusion: saves re-reading p and q).
        CPU: eager SpMV + dot (same semantics)."""
        if not self._values.is_cuda:
            self._spmv(p, out=q)
            dot = torch.dot(q.local, p.local) if q.local.numel() else \
                torch.zeros((), dtype=q.local.dtype)
            comm.all_reduce_(dot)
            return dot
        from . import kernels

        plan = self._xplan(p.partition)
        precise = not isinstance(plan, WindowGatherPlan)
        dm = self._dia() if not precise else None
        ell = None if dm is not None else (
            self._ell() if not precise else None)
        if dm is not None:
            ws_ = plan.hi - plan.lo
            if comm.world_size() > 1 and not os.environ.get("SPARSE_NO_OVERLAP"):
                h = plan.gather_halos_begin(p.local)
                a, bnd = self._dia_interior(dm, plan)
                dot = kernels.dia_spmv_dot(dm, plan.handle_pieces(h), q.local,
                                           p.local, plan.lo, ws_, a, bnd)
                pieces = plan.gather_halos_end(h)
                dot = dot + kernels.dia_spmv_dot(dm, pieces, q.local, p.local,
                                                 plan.lo, ws_, 0, a)
                dot = dot + kernels.dia_spmv_dot(dm, pieces, q.local, p.local,
                                                 plan.lo, ws_, bnd, -1)
            else:
                pieces = plan.gather_halos(p.local)
                dot = kernels.dia_spmv_dot(dm, pieces, q.local, p.local,
                                           plan.lo, ws_)
        elif ell is not None:
            a = b = 0
            if comm.world_size() > 1 and not os.environ.get("SPARSE_NO_OVERLAP"):
                a, b = self._ell_interior(ell, plan)
            if b > a:
                h = plan.gather_halos_begin(p.local)
                dot = kernels.ell_spmv_dot(ell, plan.handle_pieces(h),
                                           q.local, p.local, plan.lo, a, b)
                pieces = plan.gather_halos_end(h)
                dot = dot + kernels.ell_spmv_dot(ell, pieces, q.local,
                                                 p.local, plan.lo, 0, a)
                dot = dot + kernels.ell_spmv_dot(ell, pieces, q.local,
                                                 p.local, plan.lo, b, -1)
            else:
                pieces = plan.gather_halos(p.local)
                dot = kernels.ell_spmv_dot(ell, pieces, q.local, p.local,
                                           plan.lo)
        else:
            xw = plan.gather(p.local)
            dot = torch.zeros((), dtype=self._values.dtype, device=self._values.device)
            lc = self.local
            if precise:
                lc = ops.LocalCSR(lc.indptr, plan.remap(lc.indices),
                                  lc.values, lc.nrows, plan.hi,
                                  lc.max_row_nnz)
            kernels.spmv_dot(lc, xw, q.local, p.local, dot, plan.lo)
        comm.all_reduce_(dot)
        return dot

    def spmv_bpdot(self, r: DistArray, p_old: DistArray, p_new: DistArray,
                   q: DistArray, beta_num: torch.Tensor,
                   beta_den: torch.Tensor):
        """Fused CG K1 (DIA fast path only): p_new = r + β p_old with
        β = beta_num/beta_den read on device, q = A @ p_new, returns the
        all-reduced p_new·q.  Returns None when this matrix has no DIA
        mirror (caller falls back to the 3-kernel loop).  p_new must not
        alias p_old (double-buffered by the caller).  This is the MI355X
        two-kernel CG iteration: the separate p-update HBM pass disappears
        (reference CG task chain, linalg.py:499-565)."""
        dm = self._dia()
        if dm is None:
            return None
        from . import kernels

        plan = self._xplan(r.partition)
        r_pieces = plan.gather_halos(r.local)
        p_pieces = plan.gather_halos(p_old.local)
        dot = kernels.dia_spmv_bpdot(dm, r_pieces, p_pieces, p_new.local,
                                     q.local, beta_num, beta_den, plan.lo,
                                     plan.hi - plan.lo)
        comm.all_reduce_(dot)
        return dot

    def _rspmm(self, A: DistArray) -> DistArray:
        # C = A(dense k x m) @ self(m x n): replicate A, local partial with my
        # row slab of B, ADD all-reduce (reference csr.py:1209-1240 semantics).
        if A.shape[1] != self.shape[0]:
            raise ValueError(f"dimension mismatch {A.shape} @ {self.shape}")
        Ag = A.gather()
        vdt = self._out_dtype(Ag.dtype)
        r = comm.rank()
        Aslab = Ag[:, self.partition.start(r): self.partition.stop(r)].to(vdt)
        lc = self.local
        if lc.values.dtype != vdt:
            lc = ops.LocalCSR(lc.indptr, lc.indices, lc.values.to(vdt), lc.nrows, lc.ncols)
        C = ops.rspmm(Aslab.contiguous(), lc)  # (k, n) partial
        comm.all_reduce_(C)
        return DistArray.from_global(C)

    def _spgemm(self, B: "csr_array") -> "csr_array":
        # 1-D row algorithm (reference csr.py:1317-1490): fetch the B rows
        # my slab references, multiply locally.  Two gather modes
        # (VERDICT r1 #2 comm-scalable SpGEMM):
        #  - window: all B rows in my min/max column window (cheap plan,
        #    right for banded operands where the window IS the reference set)
        #  - precise: ship only the DISTINCT referenced rows (the row
        #    analog of PreciseGatherPlan) when they cover < 1/2 of the
        #    window — O(nnz referenced) per rank, not O(nnz window).
        from .parallel.shuffle import gather_csr_rows_precise

        if self.shape[1] != B.shape[0]:
            raise ValueError(f"dimension mismatch {self.shape} @ {B.shape}")
        lo, hi = self._col_window()
        lc = self.local
        cols = None
        if comm.world_size() > 1:
            cols = torch.unique(self._indices.to(torch.int64))
            # empty slab: either branch is fine — vote precise so it never
            # vetoes; collective-sequence safety: ALL ranks must take one
            # branch, so all-reduce(min) the per-rank vote
            use_precise = (cols.numel() == 0
                           or int(cols.numel()) < (hi - lo) // 2)
            flag = torch.tensor([1.0 if use_precise else 0.0])
            comm.all_reduce_(flag, op="min")
            if not flag.item() > 0.5:
                cols = None
        if cols is not None:
            bip, bix, bvs = gather_csr_rows_precise(
                B._indptr, B._indices, B._values, B.partition, cols)
            vdt = common_value_dtype(self._values.dtype, bvs.dtype)
            aix = torch.searchsorted(cols, lc.indices.to(torch.int64))
            A_l = ops.LocalCSR(lc.indptr, aix.to(lc.indices.dtype),
                               lc.values.to(vdt), lc.nrows, cols.numel())
            B_l = ops.LocalCSR(bip, bix, bvs.to(vdt), cols.numel(), B.shape[1])
            C_l = ops.spgemm(A_l, B_l, a_col_lo=0)
        else:
            bip, bix, bvs = gather_csr_rows(B._indptr, B._indices, B._values,
                                            B.partition, lo, hi)
            vdt = common_value_dtype(self._values.dtype, bvs.dtype)
            A_l = ops.LocalCSR(lc.indptr, lc.indices, lc.values.to(vdt),
                               lc.nrows, hi)
            B_l = ops.LocalCSR(bip, bix, bvs.to(vdt), hi - lo, B.shape[1])
            C_l = ops.spgemm(A_l, B_l, a_col_lo=lo)
        return csr_array.from_local(C_l.indptr, C_l.indices, C_l.values,
                                    self.partition, (self.shape[0], B.shape[1]))

    def _spgemm_2d(self, B) -> "csr_array":
        """C = A(csr) @ B(csc) on a 2-D (gx, gy) process grid — the MI355X
        realization of the reference's SPGEMM_CSR_CSR_CSC 2-D replicated
        algorithm (csr.py:1495-1728, spgemm_csr_csr_csc.cu): rank (pi, pj)
        gathers A's row block pi and B's column block pj (B stays CSC — no
        global transpose), multiplies locally, then shuffles tile rows to
        the 1-D row owners.  Per-rank comm is O(nnz(A)/gx + nnz(B)/gy +
        nnz(C)/W) vs the 1-D algorithm's O(nnz(B)) B-row broadcast."""
        from .parallel.shuffle import gather_csr_rows, shuffle_to_owner
        from .utils import factor_int

        if self.shape[1] != B.shape[0]:
            raise ValueError(f"dimension mismatch {self.shape} @ {B.shape}")
        W = comm.world_size()
        gy, gx = factor_int(W)  # gx >= gy: more row blocks than col blocks
        pi, pj = divmod(comm.rank(), gy)
        m, K, n = self.shape[0], self.shape[1], B.shape[1]
        rb = RowPartition.equal(m, gx)
        cb = RowPartition.equal(n, gy)
        r0, r1 = rb.start(pi), rb.stop(pi)
        c0, c1 = cb.start(pj), cb.stop(pj)
        # A row block pi (global rows [r0, r1)); every rank participates in
        # both collective gathers with its own target range
        aip, aix, avs = gather_csr_rows(self._indptr, self._indices,
                                        self._values, self.partition, r0, r1)
        # B column block pj, still column-compressed
        bip, bix, bvs = gather_csr_rows(B._colptr, B._indices, B._values,
                                        B.partition, c0, c1)
        # local CSC (K x (c1-c0)) -> CSR by stable row sort (device torch ops)
        ccols = torch.repeat_interleave(
            torch.arange(c1 - c0, dtype=torch.int64, device=bix.device),
            bip[1:] - bip[:-1])
        order = torch.argsort(bix.to(torch.int64), stable=True)
        brows = bix.to(torch.int64)[order]
        bcsr_ip = torch.zeros(K + 1, dtype=torch.int64, device=bix.device)
        if brows.numel():
            bcsr_ip[1:] = torch.cumsum(
                torch.bincount(brows, minlength=K), dim=0)
        vdt = common_value_dtype(self._values.dtype, bvs.dtype)
        A_l = ops.LocalCSR(aip, aix, avs.to(vdt), r1 - r0, K)
        B_l = ops.LocalCSR(bcsr_ip, ccols[order].to(self._indices.dtype),
                           bvs[order].to(vdt), K, c1 - c0)
        C_l = ops.spgemm(A_l, B_l, a_col_lo=0)
        # tile (r1-r0 x c1-c0) -> global COO -> shuffle rows to 1-D owners
        trows = r0 + torch.repeat_interleave(
            torch.arange(r1 - r0, dtype=torch.int64, device=aix.device),
            C_l.indptr[1:] - C_l.indptr[:-1])
        tcols = C_l.indices.to(torch.int64) + c0
        part = RowPartition.equal(m, W)
        i, j, v = shuffle_to_owner(trows, part, tcols, C_l.values)
        me = comm.rank()
        mloc = part.count(me)
        idt = torch.int32 if n < 2**31 - 1 else torch.int64
        indptr, cols_l, v = ops.local_coo_to_csr(
            i - part.start(me), j.to(idt), v.to(vdt), mloc, n)
        return csr_array.from_local(indptr, cols_l.to(idt), v,
                                    part, (m, n))

    def sddmm(self, C, D) -> "csr_array":
        """vals'[i,j] = vals[i,j] * (C[i,:] @ D[:,j]) (reference csr.py:693-705).

        Operand-block gathers (reference csr.py:1244-1312): C rows are
        fetched only for MY row slab (window gather over C's partition);
        D is fetched as the COLUMN block of my min/max column window
        (MinMaxImage proj dim 1) — per-rank D traffic = k x window, not
        k x n."""
        from .parallel.gather import ColBlockGatherPlan

        C = asdistarray(C)
        D = asdistarray(D)
        r = comm.rank()
        if C.partition == self.partition:
            Clocal = C.local
        else:
            key = ("sddmm_c", C.partition.starts)
            if key not in self._plan_cache:
                self._plan_cache[key] = WindowGatherPlan(
                    self.partition.start(r), self.partition.stop(r),
                    C.partition)
            Clocal = self._plan_cache[key].gather(C.local)
        lo, hi = self._col_window()
        key = ("sddmm_d", D.partition.starts)
        if key not in self._plan_cache:
            self._plan_cache[key] = ColBlockGatherPlan(lo, hi, D.partition)
        Dblk = self._plan_cache[key].gather(D.local)
        vdt = common_value_dtype(self._values.dtype,
                                 common_value_dtype(Clocal.dtype, Dblk.dtype))
        lc = self.local
        out = ops.sddmm(ops.LocalCSR(lc.indptr, lc.indices, lc.values.to(vdt),
                                     lc.nrows, lc.ncols),
                        Clocal.to(vdt), Dblk.to(vdt), col_lo=lo)
        return csr_array.from_local(self._indptr, self._indices, out,
                                    self.partition, self.shape)

    def tropical_spmv(self, other, out=None):
        """(max, lex-min) semiring SpMV on int64 multi-field vectors
        (reference csr.py:366-424, tropical_spmv.cu)."""
        x = asdistarray(other)
        assert x.ndim == 2
        plan = self._xplan(x.partition)
        xw = plan.gather(x.local)
        lc = self.local
        y = ops.tropical_spmv(lc, xw, col_lo=plan.lo)
        res = DistArray.from_local(y, self.partition, (self.shape[0], x.shape[1]))
        if out is not None:
            out.local.copy_(res.local)
            return out
        return res

    # -- conversions ----------------------------------------------------------
    def tocsr(self, copy=False):
        return self.copy() if copy else self

    def tocoo(self, copy=False):
        from .coo import coo_array

        rows = ops.expand_pos_to_coordinates(self._indptr, self._values.numel(),
                                             row_offset=self.partition.start(comm.rank()))
        return coo_array._from_local(rows.to(self._indices.dtype), self._indices,
                                     self._values, self.shape)

    def tocsc(self, copy=False):
        from .csc import csc_array

        # distributed transpose-exchange: send each nnz to its column owner
        cpart = RowPartition.equal(self.shape[1], comm.world_size())
        rows = ops.expand_pos_to_coordinates(self._indptr, self._values.numel(),
                                             row_offset=self.partition.start(comm.rank()))
        cols, rows2, vals = shuffle_to_owner(self._indices.to(torch.int64), cpart,
                                             rows.to(torch.int64), self._values)
        # local (col, row) grouping via the segmented scatter+sort kernel
        me = comm.rank()
        c0 = cpart.start(me)
        ncl = cpart.count(me)
        idt = index_dtype_for(self.shape)
        colptr, rows2, vals = ops.local_coo_to_csr(
            cols - c0, rows2.to(idt), vals, ncl, self.shape[0])
        return csc_array.from_local(colptr, rows2.to(idt), vals, cpart, self.shape)

    def todia(self, copy=False):
        from .dia import dia_array

        return self.tocoo().todia()

    def transpose(self, copy=False):
        """Zero-copy view: CSR(m,n).T == CSC(n,m) with columns partitioned by
        this matrix's row partition (reference csr.py:620-627)."""
        from .csc import csc_array

        if copy:
            return csc_array.from_local(self._indptr.clone(), self._indices.clone(),
                                        self._values.clone(), self.partition,
                                        (self.shape[1], self.shape[0]))
        return csc_array.from_local(self._indptr, self._indices, self._values,
                                    self.partition, (self.shape[1], self.shape[0]))

    @property
    def T(self):
        return self.transpose()

    def diagonal(self, k=0):
        if k != 0:
            m = self.tocoo()
            return m.diagonal(k=k)
        r = comm.rank()
        d_local = ops.csr_diagonal(self.local, row_offset=self.partition.start(r))
        dlen = min(self.shape)
        starts = [min(s, dlen) for s in self.partition.starts]
        part = RowPartition.from_starts(starts)
        return DistArray.from_local(d_local[: part.count(r)], part, (dlen,))

    def todense(self, order=None, out=None):
        D = ops.csr_to_dense(self.local)
        res = DistArray.from_local(D, self.partition, self.shape)
        if out is not None:
            np.copyto(out, res.numpy())
            return out
        return res

    # -- elementwise ----------------------------------------------------------
    def multiply(self, other):
        from .csc import csc_array
        from .coo import coo_array

        if isinstance(other, numbers.Number) or (
            isinstance(other, torch.Tensor) and other.dim() == 0):
            return self._with_values(lambda v: v * other)
        if isinstance(other, (csc_array, coo_array)):
            other = other.tocsr()
        if isinstance(other, csr_array):
            return self._elem_mult(other)
        # dense operand: keep the sparse structure (reference csr.py:1102-1147)
        d = asdistarray(other) if not isinstance(other, DistArray) else other
        if d.ndim == 2 and d.shape == self.shape:
            r = comm.rank()
            Dl = d.local if d.partition == self.partition else d.gather()[
                self.partition.start(r): self.partition.stop(r)]
            vdt = common_value_dtype(self._values.dtype, Dl.dtype)
            lc = self.local
            out = ops.mult_dense(ops.LocalCSR(lc.indptr, lc.indices,
                                              lc.values.to(vdt), lc.nrows, lc.ncols),
                                 Dl.to(vdt))
            return csr_array.from_local(self._indptr, self._indices, out,
                                        self.partition, self.shape)
        if d.ndim == 1 and d.shape[0] == self.shape[1]:
            # row-vector broadcast: vals *= v[col]
            vg = d.gather()
            vdt = common_value_dtype(self._values.dtype, vg.dtype)
            return self._with_values(lambda v: v.to(vdt) * vg[self._indices.long()].to(vdt))
        raise NotImplementedError("multiply with this operand shape")

    def __mul__(self, other):
        return self.multiply(other)

    __rmul__ = __mul__

    def _align(self, other: "csr_array") -> "csr_array":
        if other.partition == self.partition:
            return other
        o = other.copy()
        o._repartition(self.partition)
        return o

    def _elem_mult(self, other: "csr_array") -> "csr_array":
        if self.shape != other.shape:
            raise ValueError("inconsistent shapes")
        o = self._align(other)
        C = ops.elem_mult(self.local, o.local)
        return csr_array.from_local(C.indptr, C.indices, C.values, self.partition,
                                    self.shape)

    def _add_sub(self, other, beta) -> "csr_array":
        if self.shape != other.shape:
            raise ValueError("inconsistent shapes")
        o = self._align(other.tocsr() if not isinstance(other, csr_array) else other)
        C = ops.add(self.local, o.local, alpha=1.0, beta=beta)
        return csr_array.from_local(C.indptr, C.indices, C.values, self.partition,
                                    self.shape)

    def __add__(self, other):
        from .coo import coo_array
        from .csc import csc_array

        if isinstance(other, numbers.Number):
            if other == 0:
                return self.copy()
            raise NotImplementedError("adding a nonzero scalar to a sparse matrix")
        if isinstance(other, (csr_array, csc_array, coo_array)):
            return self._add_sub(other, +1.0)
        return self.todense() + asdistarray(other)

    __radd__ = __add__

    def __sub__(self, other):
        from .coo import coo_array
        from .csc import csc_array

        if isinstance(other, numbers.Number):
            if other == 0:
                return self.copy()
            raise NotImplementedError("subtracting a nonzero scalar from a sparse matrix")
        if isinstance(other, (csr_array, csc_array, coo_array)):
            return self._add_sub(other, -1.0)
        return self.todense() - asdistarray(other)

    def __rsub__(self, other):
        return (-self).__add__(other)

    def __truediv__(self, other):
        if isinstance(other, numbers.Number) or (
            isinstance(other, torch.Tensor) and other.dim() == 0):
            return self._with_values(lambda v: v / other)
        raise NotImplementedError("sparse division by non-scalar")

    def __str__(self):
        return str(self.to_scipy_sparse_csr())

    @classmethod
    def make_empty(cls, shape, dtype):
        return cls(tuple(shape), dtype=dtype)


csr_matrix = csr_array
