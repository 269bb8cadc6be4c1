"""MatrixMarket I/O.

Reference parity: sparse/io.py:24-51 + the C++ parser
src/sparse/io/mtx_to_coo.cc:31-135 (header/symmetry handling,
pattern/integer/real/complex fields, 1->0 based, symmetric expansion).

Byte-range-parallel ingest: every rank reads the header,
then parses ONLY its byte chunk of the data section (line-straddle handled
by the skip-first / finish-last convention), so host RAM and parse time are
O(nnz/W) per rank — the at-scale equivalent of the reference's single parse
task + runtime partitioning.  mmwrite is provided as well (the reference
has none; checkpoint parity) with a vectorized C block writer.
"""
from __future__ import annotations

import io as _io

import numpy as np

from .coo import coo_array

__all__ = ["mmread", "mmwrite"]


def _parse_chunk(buf: bytes, field: str) -> np.ndarray:
    """Parse whitespace-separated coordinate lines from a bytes block."""
    width = {"pattern": 2, "complex": 4}.get(field, 3)
    if not buf.strip():
        return np.zeros((0, width))
    try:  # pandas C reader is 10-50x numpy loadtxt
        import pandas as pd

        data = pd.read_csv(_io.BytesIO(buf), sep=r"\s+", header=None,
                           dtype=np.float64, comment="%").to_numpy()
    except ImportError:
        data = np.loadtxt(_io.BytesIO(buf), ndmin=2)
    if data.ndim == 1:
        data = data.reshape(1, -1)
    return data


def mmread(path) -> coo_array:
    from .parallel import comm

    with open(path, "rb") as f:
        header = f.readline().decode().strip()
        parts = header.split()
        if len(parts) < 5 or not parts[0].startswith("%%MatrixMarket"):
            raise ValueError(f"{path}: not a MatrixMarket file")
        _, obj, fmt, field, symmetry = parts[:5]
        obj, fmt = obj.lower(), fmt.lower()
        field, symmetry = field.lower(), symmetry.lower()
        if obj != "matrix" or fmt != "coordinate":
            raise NotImplementedError(f"mmread: {obj}/{fmt} not supported")
        line = f.readline()
        while line.startswith(b"%"):
            line = f.readline()
        m, n, nnz = (int(x) for x in line.split())
        data_start = f.tell()
        f.seek(0, 2)
        data_end = f.tell()
        ws = comm.world_size()
        me = comm.rank()
        span = data_end - data_start

        def first_line_start_at_or_after(b: int) -> int:
            """Smallest line-start byte >= b (data_end if none)."""
            if b <= data_start:
                return data_start
            if b >= data_end:
                return data_end
            f.seek(b - 1)
            if f.read(1) == b"\n":
                return b
            f.readline()  # finish the line in progress at b
            return min(f.tell(), data_end)

        # rank me owns every line whose START byte falls in [b0, b1)
        b0 = data_start + (span * me) // ws
        b1 = data_start + (span * (me + 1)) // ws
        start = first_line_start_at_or_after(b0)
        end = first_line_start_at_or_after(b1) if me != ws - 1 else data_end
        if end > start:
            f.seek(start)
            buf = f.read(end - start)
        else:
            buf = b""
    data = _parse_chunk(buf, field)
    rows = data[:, 0].astype(np.int64) - 1
    cols = data[:, 1].astype(np.int64) - 1
    if field == "pattern":
        vals = np.ones(len(rows), dtype=np.float64)
    elif field == "complex":
        vals = data[:, 2] + 1j * data[:, 3]
    else:  # real / integer
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
    # sanity: global count must match the header (+ symmetric expansion)
    import torch as _t

    cnt = _t.tensor([float(len(vals))])
    comm.all_reduce_(cnt)
    base = nnz if symmetry == "general" else None
    if base is not None and int(cnt.item()) != base:
        raise ValueError(
            f"{path}: parsed {int(cnt.item())} entries, header says {base}")
    from .runtime import runtime
    from .types import index_dtype_for

    rt = runtime()
    idt = index_dtype_for((m, n))
    import torch

    return coo_array._from_local(
        torch.as_tensor(rows, device=rt.device).to(idt),
        torch.as_tensor(cols, device=rt.device).to(idt),
        torch.as_tensor(vals, device=rt.device), (m, n))


def _format_block(rows, cols, vals, cplx: bool) -> str:
    """Vectorized coordinate-line formatting (no per-nnz Python loop)."""
    cols_d = {"r": rows + 1, "c": cols + 1}
    if cplx:
        cols_d["re"] = vals.real
        cols_d["im"] = vals.imag
    else:
        cols_d["v"] = vals
    try:
        import pandas as pd

        sio = _io.StringIO()
        pd.DataFrame(cols_d).to_csv(sio, sep=" ", header=False, index=False,
                                    float_format="%.17g")
        return sio.getvalue()
    except ImportError:
        sio = _io.StringIO()
        arr = np.column_stack([v for v in cols_d.values()])
        fmt = "%d %d " + ("%.17g %.17g" if cplx else "%.17g")
        np.savetxt(sio, arr, fmt=fmt)
        return sio.getvalue()


def mmwrite(path, A, comment: str = "") -> None:
    from .parallel import comm

    c = A.tocoo() if A.format != "coo" else A
    rows, cols, vals = c.row, c.col, c.data  # collective gathers: all ranks
    if comm.rank() == 0:
        cplx = np.iscomplexobj(vals)
        field = "complex" if cplx else "real"
        with open(path, "w") as f:
            f.write(f"%%MatrixMarket matrix coordinate {field} general\n")
            if comment:
                for ln in comment.splitlines():
                    f.write(f"%{ln}\n")
            f.write(f"{A.shape[0]} {A.shape[1]} {len(vals)}\n")
            CH = 1 << 24  # bounded formatting buffers at capacity scale
            for s in range(0, len(vals), CH):
                f.write(_format_block(rows[s: s + CH], cols[s: s + CH],
                                      vals[s: s + CH], cplx))
    if comm.initialized():
        # other ranks may mmread the file right after this returns
        import torch.distributed as dist

        dist.barrier()


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
