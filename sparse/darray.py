"""DistArray: a row-partitioned dense array over one node's GPUs.

This is the MI355X-native replacement for the cuNumeric arrays the reference
leans on for vectors/matrices: a torch tensor slab per rank +
a RowPartition, with dot/norm doing a local reduce + RCCL all-reduce, and
__array__ materializing the global numpy array (all-gather) for tests and
host-side code.

Scalars produced by reductions stay 0-dim device tensors ("futures" in the
reference's sense, linalg.py:479-496) so solver inner loops never block on
the host.
"""
from __future__ import annotations

import numbers
from typing import Optional, Union

import numpy as np
import torch

from .parallel import comm
from .parallel.partition import RowPartition
from .runtime import runtime
from .types import to_torch_dtype, to_numpy_dtype

Scalar = Union[int, float, complex, torch.Tensor]


class DistArray:
    __array_priority__ = 20.0  # beat numpy in mixed binary ops

    def __init__(self, local: torch.Tensor, partition: RowPartition, shape: tuple):
        self.local = local
        self.partition = partition
        self.gshape = tuple(int(s) for s in shape)
        assert local.shape[0] == partition.count(comm.rank()), (
            f"local slab {tuple(local.shape)} != partition count "
            f"{partition.count(comm.rank())}"
        )

    # -- construction ---------------------------------------------------------
    @staticmethod
    def from_global(data, dtype=None, partition: Optional[RowPartition] = None) -> "DistArray":
        """Every rank passes the same global array; each keeps its slab."""
        rt = runtime()
        if isinstance(data, DistArray):
            return data if dtype is None else data.astype(dtype)
        if isinstance(data, torch.Tensor):
            t = data
        else:
            t = torch.as_tensor(np.asarray(data))
        if dtype is not None:
            t = t.to(to_torch_dtype(dtype))
        shape = tuple(t.shape)
        if partition is None:
            partition = RowPartition.equal(shape[0], comm.world_size()) if shape else RowPartition.single(0)
        r = comm.rank()
        local = t[partition.start(r): partition.stop(r)].to(rt.device).contiguous()
        return DistArray(local, partition, shape)

    @staticmethod
    def from_local(local: torch.Tensor, partition: RowPartition, shape=None) -> "DistArray":
        if shape is None:
            shape = (partition.n, *local.shape[1:])
        return DistArray(local, partition, tuple(shape))

    # -- properties -----------------------------------------------------------
    @property
    def shape(self):
        return self.gshape

    @property
    def ndim(self):
        return len(self.gshape)

    @property
    def size(self):
        n = 1
        for s in self.gshape:
            n *= s
        return n

    @property
    def dtype(self):
        return to_numpy_dtype(self.local.dtype)

    @property
    def tdtype(self):
        return self.local.dtype

    @property
    def device(self):
        return self.local.device

    def __len__(self):
        return self.gshape[0]

    # -- materialization ------------------------------------------------------
    def gather(self) -> torch.Tensor:
        """Global tensor, replicated on every rank."""
        return comm.all_gather_rows(self.local, self.partition.counts())

    def numpy(self) -> np.ndarray:
        return self.gather().cpu().numpy()

    def __array__(self, dtype=None):
        a = self.numpy()
        return a.astype(dtype) if dtype is not None else a

    def __repr__(self):
        return f"DistArray(shape={self.gshape}, dtype={self.dtype})\n{self.numpy()!r}"

    # -- shape/dtype ops ------------------------------------------------------
    def astype(self, dtype) -> "DistArray":
        return DistArray(self.local.to(to_torch_dtype(dtype)), self.partition, self.gshape)

    def copy(self) -> "DistArray":
        return DistArray(self.local.clone(), self.partition, self.gshape)

    def fill(self, v) -> None:
        self.local.fill_(v)

    def conj(self) -> "DistArray":
        return DistArray(self.local.conj().resolve_conj(), self.partition, self.gshape)

    def real(self) -> "DistArray":
        return DistArray(self.local.real.contiguous(), self.partition, self.gshape)

    def reshape2d(self, k: int) -> "DistArray":
        """(n*k,) -> (n, k) when the flat partition is k-aligned; used by
        solvers that flatten multi-vectors."""
        assert self.ndim == 1 and self.gshape[0] % k == 0
        assert all(s % k == 0 for s in self.partition.starts)
        part = RowPartition.from_starts([s // k for s in self.partition.starts])
        return DistArray(self.local.view(-1, k), part, (self.gshape[0] // k, k))

    def flatten1d(self) -> "DistArray":
        assert self.ndim == 2
        k = self.gshape[1]
        part = RowPartition.from_starts([s * k for s in self.partition.starts])
        return DistArray(self.local.reshape(-1), part, (self.gshape[0] * k,))

    # -- arithmetic -----------------------------------------------------------
    def _coerce(self, other):
        if isinstance(other, DistArray):
            assert other.partition == self.partition, "partition mismatch"
            return other.local
        if isinstance(other, torch.Tensor) and other.dim() == 0:
            return other
        if isinstance(other, numbers.Number):
            return other
        if isinstance(other, (np.ndarray, list, tuple)):
            return DistArray.from_global(other, partition=self.partition).local
        return NotImplemented

    def _bin(self, other, fn):
        o = self._coerce(other)
        if o is NotImplemented:
            return NotImplemented
        out = fn(self.local, o)
        return DistArray(out, self.partition, self.gshape)

    def __add__(self, o):
        return self._bin(o, torch.add)

    __radd__ = __add__

    def __sub__(self, o):
        return self._bin(o, torch.sub)

    def __rsub__(self, o):
        return self._bin(o, lambda a, b: b - a)

    def __mul__(self, o):
        return self._bin(o, torch.mul)

    __rmul__ = __mul__

    def __truediv__(self, o):
        return self._bin(o, torch.div)

    def __rtruediv__(self, o):
        return self._bin(o, lambda a, b: b / a)

    def __pow__(self, o):
        return self._bin(o, torch.pow)

    def __neg__(self):
        return DistArray(-self.local, self.partition, self.gshape)

    def __abs__(self):
        return DistArray(torch.abs(self.local), self.partition, self.gshape)

    def __iadd__(self, o):
        c = self._coerce(o)
        self.local.add_(c if isinstance(c, torch.Tensor) else torch.as_tensor(c, device=self.local.device))
        return self

    def __isub__(self, o):
        c = self._coerce(o)
        self.local.sub_(c if isinstance(c, torch.Tensor) else torch.as_tensor(c, device=self.local.device))
        return self

    def __imul__(self, o):
        c = self._coerce(o)
        self.local.mul_(c if isinstance(c, torch.Tensor) else torch.as_tensor(c, device=self.local.device))
        return self

    def __matmul__(self, o):
        if isinstance(o, DistArray) and self.ndim == 1 and o.ndim == 1:
            return self.dot(o)
        return NotImplemented

    # -- reductions (return 0-dim device tensors: non-blocking scalars) -------
    def dot(self, other: "DistArray") -> torch.Tensor:
        o = other.local if isinstance(other, DistArray) else other
        if self.local.numel() == 0:
            out = torch.zeros((), dtype=self.local.dtype, device=self.local.device)
        elif self.local.dim() == 1 and isinstance(o, torch.Tensor) and o.dim() == 1 \
                and self.local.dtype == o.dtype:
            # fused single-pass kernel (vdot conjugates the left operand)
            out = torch.vdot(self.local, o) if self.local.is_complex() else torch.dot(self.local, o)
        elif self.local.is_complex() or (isinstance(o, torch.Tensor) and o.is_complex()):
            out = torch.sum(torch.conj(self.local) * o)
        else:
            out = torch.sum(self.local * o)
        comm.all_reduce_(out)
        return out

    def sum(self) -> torch.Tensor:
        out = torch.sum(self.local)
        comm.all_reduce_(out)
        return out

    def norm(self) -> torch.Tensor:
        if self.local.is_complex():
            out = torch.sum(torch.abs(self.local) ** 2)
        else:
            out = torch.sum(self.local * self.local)
        comm.all_reduce_(out)
        return torch.sqrt(out.real if out.is_complex() else out)

    def max(self) -> torch.Tensor:
        out = torch.max(self.local) if self.local.numel() else torch.full(
            (), -float("inf"), dtype=self.local.dtype, device=self.local.device)
        comm.all_reduce_(out, op="max")
        return out

    def amax_abs(self) -> torch.Tensor:
        out = torch.max(torch.abs(self.local)) if self.local.numel() else torch.zeros(
            (), dtype=self.local.real.dtype if self.local.is_complex() else self.local.dtype,
            device=self.local.device)
        comm.all_reduce_(out, op="max")
        return out


# -- numpy-like free functions (replicated semantics, distributed storage) ----
def _make(shape, dtype, fillfn) -> DistArray:
    rt = runtime()
    if isinstance(shape, int):
        shape = (shape,)
    part = RowPartition.equal(shape[0], comm.world_size())
    r = comm.rank()
    local = fillfn((part.count(r), *shape[1:]), to_torch_dtype(dtype), rt.device)
    return DistArray(local, part, tuple(shape))


def zeros(shape, dtype=np.float64) -> DistArray:
    return _make(shape, dtype, lambda s, d, dev: torch.zeros(s, dtype=d, device=dev))


def ones(shape, dtype=np.float64) -> DistArray:
    return _make(shape, dtype, lambda s, d, dev: torch.ones(s, dtype=d, device=dev))


def full(shape, v, dtype=np.float64) -> DistArray:
    return _make(shape, dtype, lambda s, d, dev: torch.full(s, v, dtype=d, device=dev))


def empty(shape, dtype=np.float64) -> DistArray:
    return _make(shape, dtype, lambda s, d, dev: torch.empty(s, dtype=d, device=dev))


def linspace(a, b, n, dtype=np.float64) -> DistArray:
    # computed per-slab to avoid materializing the global array
    rt = runtime()
    part = RowPartition.equal(n, comm.world_size())
    r = comm.rank()
    idx = torch.arange(part.start(r), part.stop(r), dtype=to_torch_dtype(dtype), device=rt.device)
    step = (b - a) / (n - 1) if n > 1 else 0.0
    return DistArray(a + idx * step, part, (n,))


def arange(n, dtype=np.int64) -> DistArray:
    rt = runtime()
    part = RowPartition.equal(int(n), comm.world_size())
    r = comm.rank()
    return DistArray(
        torch.arange(part.start(r), part.stop(r), dtype=to_torch_dtype(dtype), device=rt.device),
        part, (int(n),))


def random(shape, dtype=np.float64, seed: Optional[int] = None) -> DistArray:
    """Deterministic across world sizes: seeded per global row block."""
    rt = runtime()
    if isinstance(shape, int):
        shape = (shape,)
    part = RowPartition.equal(shape[0], comm.world_size())
    r = comm.rank()
    g = torch.Generator(device="cpu")
    g.manual_seed(12345 if seed is None else seed)
    full_t = torch.rand(shape, generator=g, dtype=to_torch_dtype(dtype))
    local = full_t[part.start(r): part.stop(r)].to(rt.device)
    return DistArray(local, part, tuple(shape))


def asdistarray(x, dtype=None) -> DistArray:
    if isinstance(x, DistArray):
        return x if dtype is None else x.astype(dtype)
    return DistArray.from_global(x, dtype=dtype)


def norm(x) -> torch.Tensor:
    return asdistarray(x).norm()


def dot(a, b) -> torch.Tensor:
    return asdistarray(a).dot(asdistarray(b))


def where_finite(x: DistArray) -> bool:
    ok = torch.isfinite(x.local).all().to(torch.int32)
    comm.all_reduce_(ok, op="min")
    return bool(ok.item())
