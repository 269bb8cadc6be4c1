"""Shared behavior of the compressed formats.

Reference parity: sparse/base.py — CompressedBase (nnz handling, asformat,
sum-via-matmul base.py:72-129, zero-preserving unary ufuncs base.py:148-188)
and DenseSparseBase (balance() nnz-balanced repartitioning base.py:198-282,
make_with_same_nnz_structure base.py:284-296).  The rect1 pos<->indptr
helpers of the reference (base.py:300-324) have no equivalent here: we store
plain scipy-style indptr.
"""
from __future__ import annotations

import numpy as np
import torch

from .parallel import comm
from .parallel.partition import RowPartition
from .runtime import runtime


class CompressedBase:
    # numpy must defer binary ops to our reflected implementations
    __array_priority__ = 30.0
    __array_ufunc__ = None

    @property
    def format(self):
        return self._format

    @property
    def ndim(self):
        return 2

    def getnnz(self):
        return self.nnz

    def get_shape(self):
        return self.shape

    def asformat(self, format, copy=False):
        if format is None or format == self.format:
            return self.copy() if copy else self
        try:
            conv = getattr(self, "to" + format)
        except AttributeError:
            raise ValueError(f"Format {format} is unknown.")
        return conv(copy=copy) if "copy" in conv.__code__.co_varnames else conv()

    def toarray(self, order=None, out=None):
        return self.todense(order=order, out=out)

    def conjugate(self, copy=True):
        return self.conj(copy=copy)

    def count_nonzero(self):
        v = self._values_tensor()
        local = int(torch.count_nonzero(v).item()) if v.numel() else 0
        t = torch.tensor([local], dtype=torch.int64)
        comm.all_reduce_(t)
        return int(t.item())

    # sum via matmul with a ones vector (reference base.py:72-129)
    def sum(self, axis=None, dtype=None, out=None):
        from . import darray

        if axis not in (None, 0, 1, -1, -2):
            raise ValueError("axis out of range")
        csr = self.tocsr() if self.format != "csr" else self
        if axis is None:
            v = csr._values_tensor()
            s = torch.sum(v) if v.numel() else torch.zeros((), dtype=csr._values_tensor().dtype)
            comm.all_reduce_(s)
            r = s.item()
            return r if dtype is None else np.dtype(dtype).type(r)
        if axis in (1, -1):
            ones = darray.ones((self.shape[1],), dtype=csr.dtype)
            res = csr.dot(ones)
        else:
            ones = darray.ones((self.shape[0],), dtype=csr.dtype)
            res = csr.T.dot(ones)
        if dtype is not None:
            res = res.astype(dtype)
        if out is not None:
            np.copyto(out, np.asarray(res))
            return out
        return res

    def mean(self, axis=None, dtype=None, out=None):
        denom = (
            self.shape[0] * self.shape[1]
            if axis is None
            else (self.shape[1] if axis in (1, -1) else self.shape[0])
        )
        s = self.sum(axis=axis, dtype=dtype, out=out)
        return s / denom

    # structures are always column-sorted and duplicate-free here (owner
    # shuffles sort and sum on construction): scipy API compat no-ops
    has_sorted_indices = True
    has_canonical_format = True

    def sort_indices(self):
        return None

    def sorted_indices(self):
        return self.copy()

    def sum_duplicates(self):
        return None

    def eliminate_zeros(self):
        """Drop stored zeros (in place), scipy semantics."""
        v = self._values_tensor()
        if v.numel() == 0:
            return
        keep = v != 0
        if bool(keep.all()):
            return
        self._eliminate_zeros_impl(keep)

    def _eliminate_zeros_impl(self, keep):
        raise NotImplementedError(
            f"eliminate_zeros not supported for {type(self).__name__}")

    def _values_tensor(self) -> torch.Tensor:
        raise NotImplementedError

    def _with_values(self, fn):
        """New array, same structure, values = fn(values)."""
        raise NotImplementedError


# Zero-preserving unary ufuncs (reference base.py:148-188): f(0) == 0 so they
# apply to the stored values only.
_UNARY_TORCH = {
    "abs": torch.abs,
    "sqrt": torch.sqrt,
    "sin": torch.sin,
    "tan": torch.tan,
    "arcsin": torch.arcsin,
    "arctan": torch.arctan,
    "sinh": torch.sinh,
    "tanh": torch.tanh,
    "arcsinh": torch.arcsinh,
    "arctanh": torch.arctanh,
    "expm1": torch.expm1,
    "log1p": torch.log1p,
    "sign": torch.sign,
    "rint": torch.round,
    "trunc": torch.trunc,
    "floor": torch.floor,
    "ceil": torch.ceil,
    "deg2rad": torch.deg2rad,
    "rad2deg": torch.rad2deg,
}

def _abs_dunder(self):
    return self._with_values(torch.abs)


CompressedBase.__abs__ = _abs_dunder

for _name, _fn in _UNARY_TORCH.items():

    def _method(self, _fn=_fn):
        return self._with_values(_fn)

    _method.__name__ = _name
    setattr(CompressedBase, _name, _method)
del _name, _fn, _method


class DenseSparseBase:
    """Formats with a rows/cols-compressed axis that can be rebalanced."""

    def balance(self):
        """Repartition rows so each rank owns ~equal nnz (reference
        base.py:198-282: preimage of an equal-nnz tiling + disjointness
        fix-up).  In-place.  No-op at world size 1."""
        if comm.world_size() == 1:
            return
        counts = self._local_row_nnz()  # torch int64, local rows
        all_counts = comm.all_gather_rows(
            counts.cpu(), self.partition.counts()
        ).numpy()
        newpart = RowPartition.balanced_from_counts(all_counts, comm.world_size())
        self._repartition(newpart)

    @classmethod
    def make_with_same_nnz_structure(cls, mat, arg, shape=None, dtype=None):
        """New array sharing mat's structure tensors with new values
        (reference base.py:284-296).  arg is the new values: a global numpy
        array (sliced to this rank's nnz), a local torch tensor, or the
        reference-style (vals, crd, pos) tuple (only vals is taken — the
        structure comes from mat)."""
        import numpy as _np

        vals = arg[0] if isinstance(arg, tuple) else arg
        old = mat._values_tensor()
        if isinstance(vals, torch.Tensor) and vals.numel() == old.numel():
            local = vals.to(old.device)
        else:
            a = _np.asarray(vals)
            counts = mat._nnz_counts()
            off = sum(counts[: comm.rank()])
            local = torch.as_tensor(
                _np.ascontiguousarray(a[off: off + counts[comm.rank()]]),
                device=old.device)
        if dtype is not None:
            from .types import to_torch_dtype

            local = local.to(to_torch_dtype(dtype))
        out = mat._with_values(lambda _v: local)
        if shape is not None and tuple(shape) != tuple(mat.shape):
            out.shape = tuple(int(x) for x in shape)
        return out
