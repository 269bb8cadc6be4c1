"""MatrixMarket I/O.

Reference parity: sparse/io.py:24-51 + the C++ parser
src/sparse/io/mtx_to_coo.cc:31-135 (header/symmetry handling,
pattern/integer/real/complex fields, 1->0 based, symmetric expansion).

Every rank parses the (node-local) file and keeps its nnz chunk — the SPMD
equivalent of the reference's single parse task + partitioning.  mmwrite is
provided as well (the reference has none; checkpoint parity).
"""
from __future__ import annotations

import numpy as np

from .coo import coo_array

__all__ = ["mmread", "mmwrite"]


def mmread(path) -> coo_array:
    with open(path, "r") as f:
        header = f.readline().strip()
        parts = header.split()
        if len(parts) < 5 or not parts[0].startswith("%%MatrixMarket"):
            raise ValueError(f"{path}: not a MatrixMarket file")
        _, obj, fmt, field, symmetry = parts[:5]
        obj, fmt = obj.lower(), fmt.lower()
        field, symmetry = field.lower(), symmetry.lower()
        if obj != "matrix" or fmt != "coordinate":
            raise NotImplementedError(f"mmread: {obj}/{fmt} not supported")
        line = f.readline()
        while line.startswith("%"):
            line = f.readline()
        m, n, nnz = (int(x) for x in line.split())
        if nnz == 0:
            data = np.zeros((0, 3))
        else:
            try:  # pandas C reader is 10-50x numpy loadtxt
                import pandas as pd

                data = pd.read_csv(f, sep=r"\s+", header=None, nrows=nnz,
                                   dtype=np.float64).to_numpy()
            except ImportError:
                data = np.loadtxt(f, max_rows=nnz, ndmin=2)
        if data.ndim == 1:
            data = data.reshape(1, -1)
    rows = data[:, 0].astype(np.int64) - 1
    cols = data[:, 1].astype(np.int64) - 1
    if field == "pattern":
        vals = np.ones(len(rows), dtype=np.float64)
    elif field == "complex":
        vals = data[:, 2] + 1j * data[:, 3]
    elif field == "integer":
        vals = data[:, 2].astype(np.float64)
    else:
        vals = data[:, 2].astype(np.float64)
    if symmetry in ("symmetric", "skew-symmetric", "hermitian"):
        off = rows != cols
        r2, c2, v2 = cols[off], rows[off], vals[off]
        if symmetry == "skew-symmetric":
            v2 = -v2
        elif symmetry == "hermitian":
            v2 = np.conj(v2)
        rows = np.concatenate([rows, r2])
        cols = np.concatenate([cols, c2])
        vals = np.concatenate([vals, v2])
    return coo_array((vals, (rows, cols)), shape=(m, n))


def mmwrite(path, A, comment: str = "") -> None:
    from .parallel import comm

    c = A.tocoo() if A.format != "coo" else A
    rows, cols, vals = c.row, c.col, c.data
    if comm.rank() != 0:
        return
    cplx = np.iscomplexobj(vals)
    field = "complex" if cplx else "real"
    with open(path, "w") as f:
        f.write(f"%%MatrixMarket matrix coordinate {field} general\n")
        if comment:
            for ln in comment.splitlines():
                f.write(f"%{ln}\n")
        f.write(f"{A.shape[0]} {A.shape[1]} {len(vals)}\n")
        for r, cc, v in zip(rows, cols, vals):
            if cplx:
                f.write(f"{r + 1} {cc + 1} {v.real:.17g} {v.imag:.17g}\n")
            else:
                f.write(f"{r + 1} {cc + 1} {v:.17g}\n")


def save_npz(file, matrix, compressed=True):
    """Save a sparse array in scipy .npz format (scipy.sparse.save_npz
    parity — checkpoint/restore superset of the reference, which has
    read-only mmread).  Gathers to the host on rank 0."""
    import scipy.sparse as _sp

    from .parallel import comm as _c

    s = matrix.tocsr().to_scipy_sparse_csr()
    if _c.rank() == 0:
        _sp.save_npz(file, s, compressed=compressed)
    if _c.initialized():
        import torch.distributed as dist

        dist.barrier()


def load_npz(file):
    """Load a scipy .npz sparse file as a distributed csr_array."""
    import scipy.sparse as _sp

    from .csr import csr_array

    return csr_array(_sp.load_npz(file).tocsr())
