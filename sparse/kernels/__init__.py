"""HIP (gfx950) kernel extension: loading and high-level wrappers.

The extension is built IN-TREE (sparse/kernels/_build/_sparse_hip.so) so the
.so travels to GPU boxes with the repo snapshot.  Build with
`python -m sparse.kernels.build` or __graft_entry__.build().

Policy: on a GPU, ops REQUIRE this extension — require() raises if it is
missing.  There is no eager/PyTorch fallback on the GPU path.
"""
from __future__ import annotations

import os

import torch

_EXT = None
_TRIED = False

_DIR = os.path.dirname(os.path.abspath(__file__))
BUILD_DIR = os.path.join(_DIR, "_build")
SO_PATH = os.path.join(BUILD_DIR, "sparse_hip.so")


def ext():
    """Return the loaded extension module, or None if unavailable."""
    global _EXT, _TRIED
    if _EXT is not None or _TRIED:
        return _EXT
    _TRIED = True
    if os.path.exists(SO_PATH):
        torch.ops.load_library(SO_PATH)
        _EXT = torch.ops.sparse_hip
    return _EXT


def require():
    if ext() is None:
        raise RuntimeError(
            "sparse: GPU op invoked but the HIP extension is not built "
            f"(expected {SO_PATH}). Build it with `python -m sparse.kernels.build`."
        )


# -- high-level wrappers ------------------------------------------------------
def spmv(A, x, y, col_lo: int, beta: float):
    # NOTE: a thread-per-row variant (csr_row_spmv) was measured 2.8x SLOWER
    # than the nnz-split kernel on scattered-column short rows (its win in
    # tools/spmv_bench.hip was banded-specific, where the ELL path applies
    # anyway) — nnz-split stays the general fallback.
    ext().spmv(A.indptr, A.indices, A.values, x, y, int(col_lo), float(beta))


def spmv_dot(A, x, y, p, dot_out, col_lo: int):
    """Fused y = A@x(window) ; dot_out += sum(p_local * y_local)."""
    ext().spmv_dot(A.indptr, A.indices, A.values, x, y, p, dot_out, int(col_lo))


def axpby_norm2(y, x, a, b, isalpha, negate):
    """Fused axpby; returns sum(y_new^2) (real dtypes) as a 0-dim tensor."""
    n = y.numel()
    if n == 0:
        # the HIP launcher returns early on n==0 without writing `partial`;
        # an empty slab must contribute an exact 0 to the all-reduced dot
        return torch.zeros((), dtype=y.dtype, device=y.device)
    ept = max(1, 16 // y.element_size())
    blocks = (n // ept + 255) // 256 + 1
    partial = torch.empty(blocks, dtype=y.dtype, device=y.device)
    ext().axpby_norm2(y, x, a, b, bool(isalpha), bool(negate), partial)
    return partial.sum()


def spmm(A, B, C, col_lo: int):
    ext().spmm(A.indptr, A.indices, A.values, B, C, int(col_lo))


def rspmm(B, A_dense, C):
    ext().rspmm(B.indptr, B.indices, B.values, A_dense, C)


def sddmm(A, C, D, out, col_lo: int = 0):
    ext().sddmm(A.indptr, A.indices, A.values, C, D, out, int(col_lo))


def mult_dense(A, D, out):
    ext().mult_dense(A.indptr, A.indices, A.values, D, out)


def csr_to_dense(A, out):
    ext().csr_to_dense(A.indptr, A.indices, A.values, out)


def dense_to_csr(D, indptr_or_counts, indices, vals, fill: bool):
    """Two-phase dense->CSR: fill=False counts nnz/row, fill=True emits
    ordered (indices, vals) via ballot compaction (dense_to_csr.cu parity)."""
    ext().dense_to_csr(D, indptr_or_counts, indices, vals, bool(fill))


def coo_to_csr(rows, cols, vals, cursor, indptr, out_idx, out_vals, flags):
    """Segmented COO->CSR: atomic scatter + per-row LDS sort; flags[0]=row
    overflow, flags[1]=duplicate columns (caller falls back)."""
    ext().coo_to_csr(rows, cols, vals, cursor, indptr, out_idx, out_vals,
                     flags)


def csr_diagonal(A, out, row_offset: int):
    ext().csr_diagonal(A.indptr, A.indices, A.values, out, int(row_offset))


def csc_spmv(colptr, rowidx, values, x, y, rlo: int):
    ext().csc_spmv(colptr, rowidx, values, x, y, int(rlo))


def csc_spmm(colptr, rowidx, values, B, C, rlo: int):
    ext().csc_spmm(colptr, rowidx, values, B, C, int(rlo))


def tropical_spmv(A, x, y, col_lo: int):
    # semiring uses structure only (reference tropical_spmv.cu:26-56)
    ext().tropical_spmv(A.indptr, A.indices, x, y, int(col_lo))


def add_csr(A, B, alpha, beta, vdt):
    from ..ops.local import LocalCSR

    nnz_per_row = torch.empty(A.nrows, dtype=torch.int64, device=A.device)
    ext().add_nnz(A.indptr, A.indices, B.indptr, B.indices, nnz_per_row)
    indptr = torch.zeros(A.nrows + 1, dtype=torch.int64, device=A.device)
    torch.cumsum(nnz_per_row, 0, out=indptr[1:])
    nnz = int(indptr[-1].item())
    indices = torch.empty(nnz, dtype=A.indices.dtype, device=A.device)
    values = torch.empty(nnz, dtype=vdt, device=A.device)
    ext().add_compute(A.indptr, A.indices, A.values.to(vdt),
                      B.indptr, B.indices, B.values.to(vdt),
                      indptr, indices, values, float(alpha), float(beta))
    return LocalCSR(indptr, indices, values, A.nrows, A.ncols)


def elem_mult_csr(A, B, vdt):
    from ..ops.local import LocalCSR

    nnz_per_row = torch.empty(A.nrows, dtype=torch.int64, device=A.device)
    ext().mult_nnz(A.indptr, A.indices, B.indptr, B.indices, nnz_per_row)
    indptr = torch.zeros(A.nrows + 1, dtype=torch.int64, device=A.device)
    torch.cumsum(nnz_per_row, 0, out=indptr[1:])
    nnz = int(indptr[-1].item())
    indices = torch.empty(nnz, dtype=A.indices.dtype, device=A.device)
    values = torch.empty(nnz, dtype=vdt, device=A.device)
    ext().mult_compute(A.indptr, A.indices, A.values.to(vdt),
                       B.indptr, B.indices, B.values.to(vdt),
                       indptr, indices, values)
    return LocalCSR(indptr, indices, values, A.nrows, A.ncols)


def spgemm_csr(A, B, a_col_lo, vdt):
    """Two-phase size-binned Gustavson SpGEMM.

    Rows are binned by product upper bound (sum of touched B-row sizes)
    into LDS hash sizes 64/256/1024/2048; rows beyond that try a CHECKED
    4096-entry table (dense-product rows usually have few distinct
    columns) and only true overflows take the expand-sort-reduce
    fallback."""
    from ..ops.local import LocalCSR

    dev = A.device
    m = A.nrows
    if m == 0 or A.nnz == 0:
        return LocalCSR(torch.zeros(m + 1, dtype=torch.int64, device=dev),
                        torch.zeros(0, dtype=A.indices.dtype, device=dev),
                        torch.zeros(0, dtype=vdt, device=dev), m, B.ncols)
    acounts = A.indptr[1:] - A.indptr[:-1]
    acols = A.indices.long() - a_col_lo
    bcounts = B.indptr[1:] - B.indptr[:-1]
    # per-row product upper bound as a SEGMENT sum (cumsum + indptr
    # gather): rows_of_nnz + index_add_ was an atomic scatter over nnz
    # elements — 79 ms/call on the 931M-nnz 3-D GMG Galerkin setup
    cs = torch.zeros(A.nnz + 1, dtype=torch.int64, device=dev)
    torch.cumsum(bcounts[acols], 0, out=cs[1:])
    ub = cs[A.indptr[1:]] - cs[A.indptr[:-1]]
    counts = torch.zeros(m, dtype=torch.int64, device=dev)
    bins = [((ub <= 32), 64), ((ub > 32) & (ub <= 128), 256),
            ((ub > 128) & (ub <= 512), 1024),
            ((ub > 512) & (ub <= 1024), 2048)]
    rowlists = [mask.nonzero(as_tuple=False).flatten() for mask, _ in bins]
    Av = A.values.to(vdt)
    Bv = B.values.to(vdt)
    for rl, (_, H) in zip(rowlists, bins):
        if rl.numel():
            ext().spgemm_nnz(A.indptr, A.indices, B.indptr, B.indices, rl,
                             counts, int(a_col_lo), H)
    # checked speculative bin: dense-PRODUCT rows usually have far fewer
    # distinct columns than products; try the 4096-entry LDS table first
    # (overflow -> counts[r] = -1 -> ESC for just those rows).  8-byte
    # value types only (complex accs exceed the 160 KB LDS budget).
    big = (ub > 1024).nonzero(as_tuple=False).flatten()
    checked_ok = torch.zeros(0, dtype=torch.int64, device=dev)
    if big.numel() and vdt.itemsize <= 8:
        ext().spgemm_nnz(A.indptr, A.indices, B.indptr, B.indices, big,
                         counts, int(a_col_lo), 4096)
        over = counts[big] < 0
        esc_rows = big[over]
        checked_ok = big[~over]
        counts[esc_rows] = 0
    else:
        esc_rows = big
    esc_sub = None
    if esc_rows.numel():
        sc = acounts[esc_rows]
        sub_ip = torch.zeros(esc_rows.numel() + 1, dtype=torch.int64, device=dev)
        torch.cumsum(sc, 0, out=sub_ip[1:])
        tot = int(sub_ip[-1].item())
        pos = (torch.arange(tot, dtype=torch.int64, device=dev)
               - torch.repeat_interleave(sub_ip[:-1], sc)
               + torch.repeat_interleave(A.indptr[esc_rows], sc))
        subA = LocalCSR(sub_ip, A.indices[pos], Av[pos], esc_rows.numel(), A.ncols)
        esc_sub = _spgemm_dense_rows(subA, B, a_col_lo, vdt)
        if esc_sub is None:
            esc_sub = _spgemm_esc(subA, B, a_col_lo, vdt)
        counts[esc_rows] = esc_sub.indptr[1:] - esc_sub.indptr[:-1]
    indptr = torch.zeros(m + 1, dtype=torch.int64, device=dev)
    torch.cumsum(counts, 0, out=indptr[1:])
    nnz = int(indptr[-1].item())
    indices = torch.empty(nnz, dtype=A.indices.dtype, device=dev)
    values = torch.empty(nnz, dtype=vdt, device=dev)
    for rl, (_, H) in zip(rowlists, bins):
        if rl.numel():
            ext().spgemm_compute(A.indptr, A.indices, Av, B.indptr, B.indices,
                                 Bv, rl, indptr, indices, values,
                                 int(a_col_lo), H)
    if checked_ok.numel():
        ext().spgemm_compute(A.indptr, A.indices, Av, B.indptr, B.indices,
                             Bv, checked_ok, indptr, indices, values,
                             int(a_col_lo), 4096)
    if esc_sub is not None and esc_sub.nnz:
        ec = counts[esc_rows]
        eoff = torch.zeros(esc_rows.numel(), dtype=torch.int64, device=dev)
        torch.cumsum(ec[:-1], 0, out=eoff[1:])
        tot = int(ec.sum().item())
        dst = (torch.arange(tot, dtype=torch.int64, device=dev)
               - torch.repeat_interleave(eoff, ec)
               + torch.repeat_interleave(indptr[esc_rows], ec))
        indices[dst] = esc_sub.indices
        values[dst] = esc_sub.values
    # rows are emitted sorted (in-kernel LDS bitonic compaction)
    return LocalCSR(indptr, indices, values, m, B.ncols)


_ESC_LIMIT = 1 << 28  # max materialized products per ESC batch
_DENSE_FALLBACK_BUDGET = 1 << 30  # max dense-accumulator elements


def _spgemm_dense_rows(A, B, a_col_lo, vdt):
    """Dense-workspace SpGEMM for a FEW very dense rows (> 2048 distinct
    output columns): scatter-add products into an (nrows, ncols) dense
    buffer, then compact.  The GPU analog of the reference's CPU dense
    workspace (spgemm_csr_csr_csr.cc:27-85); returns None when the buffer
    would exceed the budget (caller falls back to ESC)."""
    from ..ops.local import LocalCSR

    dev = A.device
    if A.nrows * B.ncols > _DENSE_FALLBACK_BUDGET:
        return None
    acc = torch.zeros(A.nrows * B.ncols, dtype=vdt, device=dev)
    touched = torch.zeros(A.nrows * B.ncols, dtype=torch.bool, device=dev)
    acounts = A.indptr[1:] - A.indptr[:-1]
    arows = torch.repeat_interleave(
        torch.arange(A.nrows, dtype=torch.int64, device=dev), acounts)
    acols = A.indices.long() - a_col_lo
    bstart = B.indptr[acols]
    bcounts = B.indptr[acols + 1] - bstart
    CH = _ESC_LIMIT
    cum = torch.cumsum(bcounts, 0)
    lo_nnz = 0
    while lo_nnz < acols.numel():
        base = int(cum[lo_nnz - 1].item()) if lo_nnz else 0
        hi_nnz = int(torch.searchsorted(cum, base + CH, right=True).item())
        hi_nnz = max(hi_nnz, lo_nnz + 1)
        bc = bcounts[lo_nnz:hi_nnz]
        tot = int(bc.sum().item())
        offs = torch.zeros(bc.numel(), dtype=torch.int64, device=dev)
        torch.cumsum(bc[:-1], 0, out=offs[1:])
        pos = (torch.arange(tot, dtype=torch.int64, device=dev)
               - torch.repeat_interleave(offs, bc)
               + torch.repeat_interleave(bstart[lo_nnz:hi_nnz], bc))
        er = torch.repeat_interleave(arows[lo_nnz:hi_nnz], bc)
        ev = (torch.repeat_interleave(A.values[lo_nnz:hi_nnz].to(vdt), bc)
              * B.values[pos].to(vdt))
        cell = er * B.ncols + B.indices[pos].long()
        acc.index_add_(0, cell, ev)
        touched[cell] = True  # structural nnz (cancellation keeps the entry)
        lo_nnz = hi_nnz
    acc2 = acc.view(A.nrows, B.ncols)
    nzmask = touched.view(A.nrows, B.ncols)
    counts = nzmask.sum(dim=1)
    indptr = torch.zeros(A.nrows + 1, dtype=torch.int64, device=dev)
    torch.cumsum(counts, 0, out=indptr[1:])
    nz = nzmask.nonzero(as_tuple=True)
    return LocalCSR(indptr, nz[1].to(A.indices.dtype), acc2[nz],
                    A.nrows, B.ncols)


def _spgemm_esc(A, B, a_col_lo, vdt):
    """Expansion-sort-compress SpGEMM fallback in torch ops (still on-GPU).
    Row-batched: the expansion materializes every intermediate product, so
    unbatched dense-product chains (e.g. repeated A@A squaring) would
    allocate tens of GB and abort; batches are capped at ~2^28 products."""
    from ..ops.local import LocalCSR

    dev = A.device
    LIMIT = _ESC_LIMIT
    acols_all = A.indices.long() - a_col_lo
    bc_all = B.indptr[acols_all + 1] - B.indptr[acols_all]
    total_all = int(bc_all.sum().item())
    if total_all > LIMIT and A.nrows > 1:
        acounts = A.indptr[1:] - A.indptr[:-1]
        arows_all = torch.repeat_interleave(
            torch.arange(A.nrows, dtype=torch.int64, device=dev), acounts)
        rowprod = torch.zeros(A.nrows, dtype=torch.int64, device=dev)
        rowprod.index_add_(0, arows_all, bc_all)
        cum = torch.cumsum(rowprod, 0).cpu().numpy()
        pieces = []
        r0 = 0
        base = 0
        import numpy as _np

        while r0 < A.nrows:
            r1 = int(_np.searchsorted(cum, base + LIMIT, side="right"))
            r1 = max(r1, r0 + 1)
            ip = (A.indptr[r0: r1 + 1] - A.indptr[r0]).contiguous()
            lo, hi = int(A.indptr[r0].item()), int(A.indptr[r1].item())
            sub = LocalCSR(ip, A.indices[lo:hi], A.values[lo:hi],
                           r1 - r0, A.ncols)
            pieces.append(_spgemm_esc(sub, B, a_col_lo, vdt))
            base = cum[r1 - 1]
            r0 = r1
        indptr = torch.zeros(A.nrows + 1, dtype=torch.int64, device=dev)
        off = 0
        chunks_i, chunks_v = [], []
        r0 = 0
        for pc in pieces:
            indptr[r0 + 1: r0 + pc.nrows + 1] = pc.indptr[1:] + off
            off += int(pc.indptr[-1].item())
            r0 += pc.nrows
            chunks_i.append(pc.indices)
            chunks_v.append(pc.values)
        return LocalCSR(indptr, torch.cat(chunks_i), torch.cat(chunks_v),
                        A.nrows, B.ncols)
    acols = acols_all
    acounts = A.indptr[1:] - A.indptr[:-1]
    arows = torch.repeat_interleave(
        torch.arange(A.nrows, dtype=torch.int64, device=dev), acounts)
    bstart = B.indptr[acols]
    bcounts = B.indptr[acols + 1] - bstart
    total = int(bcounts.sum().item())
    if total == 0:
        return LocalCSR(torch.zeros(A.nrows + 1, dtype=torch.int64, device=dev),
                        torch.zeros(0, dtype=A.indices.dtype, device=dev),
                        torch.zeros(0, dtype=vdt, device=dev), A.nrows, B.ncols)
    offs = torch.zeros(acols.numel(), dtype=torch.int64, device=dev)
    torch.cumsum(bcounts[:-1], 0, out=offs[1:])
    pos = (torch.arange(total, dtype=torch.int64, device=dev)
           - torch.repeat_interleave(offs, bcounts)
           + torch.repeat_interleave(bstart, bcounts))
    erows = torch.repeat_interleave(arows, bcounts)
    ecols = B.indices[pos].long()
    evals = torch.repeat_interleave(A.values.to(vdt), bcounts) * B.values[pos].to(vdt)
    key = erows * B.ncols + ecols
    key, order = torch.sort(key)
    evals = evals[order]
    ukey, inv = torch.unique_consecutive(key, return_inverse=True)
    out_vals = torch.zeros(ukey.numel(), dtype=vdt, device=dev)
    out_vals.index_add_(0, inv, evals)
    rows = torch.div(ukey, B.ncols, rounding_mode="floor")
    cols = (ukey - rows * B.ncols).to(A.indices.dtype)
    counts = torch.bincount(rows, minlength=A.nrows)
    indptr = torch.zeros(A.nrows + 1, dtype=torch.int64, device=dev)
    torch.cumsum(counts, 0, out=indptr[1:])
    return LocalCSR(indptr, cols, out_vals, A.nrows, B.ncols)


def axpby(y, x, a, b, isalpha: bool, negate: bool):
    """Fused y = y + (±a/b) x  (isalpha) or y = x + (±a/b) y.
    Reference: AXPBY task (axpby.cu:25-42, linalg.py:479-496)."""
    ext().axpby(y, x, a, b, bool(isalpha), bool(negate))


def rk_calc_dy(K, a, h, dy):
    """dy[i] = h * sum_j K[j,i] * a[j] (reference runge_kutta.cu:26-42)."""
    ext().rk_calc_dy(K, a, float(h), dy)


def cdist(XA, XB, out):
    ext().cdist(XA, XB, out)


# -- device-DIA fast path -----------------------------------------------------
class DiaMirror:
    """Padded diagonal planes of a banded CSR slab: values only, no index
    stream (12 -> 8 B/nnz for fp64+int32 vs ELL)."""

    __slots__ = ("dvals", "offs", "W", "m", "row0", "off_min", "off_max")

    def __init__(self, dvals, offs, W, m, row0, off_min=0, off_max=0):
        self.off_min = off_min
        self.off_max = off_max
        self.dvals = dvals
        self.offs = offs
        self.W = W
        self.m = m
        self.row0 = row0


def build_dia(A, row0: int):
    """Build the diagonal mirror of a LocalCSR (cols are GLOBAL; diagonals
    are col - global_row), or None when not banded enough."""
    m = A.nrows
    if m == 0 or A.nnz == 0:
        return None
    counts = A.indptr[1:] - A.indptr[:-1]
    rows = torch.repeat_interleave(
        torch.arange(row0, row0 + m, dtype=torch.int64, device=A.device), counts)
    diag = A.indices.long() - rows
    # chunked: torch.unique / index_put hit CUB 2^31-element limits at
    # capacity scale (3B nnz on one 288 GB GPU)
    CH = 1 << 30
    if diag.numel() <= CH:
        offs = torch.unique(diag)
    else:
        parts = [torch.unique(diag[i: i + CH])
                 for i in range(0, diag.numel(), CH)]
        offs = torch.unique(torch.cat(parts))
    W = int(offs.numel())
    mp = (m + 1) // 2 * 2
    if W == 0 or W > 48 or W * mp > 1.6 * A.nnz + 4096:
        return None
    need = W * mp * A.values.element_size()
    free, _t = torch.cuda.mem_get_info(A.values.device)
    if need > 0.5 * free:
        return None
    dvals = torch.zeros(W * mp, dtype=A.values.dtype, device=A.device)
    # one-pass scatter kernel (per-row walk) — the torch index_put_ here
    # was an nnz-scale indexFuncLargeIndex (~79 ms/call at 931M nnz)
    ext().build_dia(A.indptr, A.indices, A.values, offs.contiguous(), dvals,
                    W, int(row0))
    return DiaMirror(dvals, offs, W, m, row0,
                     off_min=int(offs[0].item()), off_max=int(offs[-1].item()))


def dia_spmv(dm: DiaMirror, pieces, y, col_lo: int, wsize: int,
             rbase: int = 0, rhi: int = -1):
    """rbase/rhi select a row sub-range (rbase even; -1 = all rows): the
    interior/boundary split that overlaps halo exchange with interior
    compute at ws>1."""
    hlo, own, hhi = pieces
    ext().dia_spmv(dm.dvals, dm.offs, hlo.contiguous(), own.contiguous(),
                   hhi.contiguous(), y, dm.W, dm.m, int(col_lo), dm.row0,
                   int(wsize), int(rbase), int(rhi))


def dia_spmv_dot(dm: DiaMirror, pieces, y, p, col_lo: int, wsize: int,
                 rbase: int = 0, rhi: int = -1):
    hlo, own, hhi = pieces
    mp = dm.dvals.numel() // dm.W
    hi = mp if rhi < 0 else rhi
    nblocks = max(1, ((hi - rbase) // 2 + 256) // 256)
    partial = torch.empty(nblocks, dtype=dm.dvals.dtype, device=dm.dvals.device)
    if hi <= rbase:
        return partial[:1].zero_().sum()
    ext().dia_spmv_dot(dm.dvals, dm.offs, hlo.contiguous(), own.contiguous(),
                       hhi.contiguous(), y, p, partial, dm.W, dm.m,
                       int(col_lo), dm.row0, int(wsize), int(rbase), int(hi))
    return partial.sum()


def dia_spmv_bpdot(dm: DiaMirror, r_pieces, p_pieces, pnew, q,
                   beta_num, beta_den, col_lo: int, wsize: int):
    """Fused CG K1: pnew = r + (beta_num/beta_den)*p_old; q = A@pnew;
    returns p.q (device 0-dim, local partial-sum).  r_pieces/p_pieces are
    the (hlo, own, hhi) window pieces of r and p_old; pnew must be a
    distinct buffer from p_old (double-buffered caller)."""
    rlo, rown, rhi = r_pieces
    plo, pown, phi = p_pieces
    mp = dm.dvals.numel() // dm.W
    nblocks = (mp // 2 + 255) // 256
    partial = torch.empty(nblocks, dtype=dm.dvals.dtype, device=dm.dvals.device)
    ext().dia_spmv_bpdot(dm.dvals, dm.offs, rlo.contiguous(), rown.contiguous(),
                         rhi.contiguous(), plo.contiguous(), pown.contiguous(),
                         phi.contiguous(), pnew, q, beta_num, beta_den,
                         partial, dm.W, dm.m, int(col_lo), dm.row0, int(wsize))
    return partial.sum()


def cg_xr_norm2(x, p, r, q, a, b):
    """Fused CG K2: x += (a/b)p; r -= (a/b)q; returns local sum(r_new^2)
    as a device 0-dim tensor."""
    n = x.numel()
    if n == 0:
        return torch.zeros((), dtype=x.dtype, device=x.device)
    ept = max(1, 16 // x.element_size())
    blocks = (n // ept + 255) // 256 + 1
    partial = torch.empty(blocks, dtype=x.dtype, device=x.device)
    ext().cg_xr_norm2(x, p, r, q, a.to(x.dtype), b.to(x.dtype), partial)
    return partial.sum()


def dia_residual(dm: DiaMirror, pieces, b, y, col_lo: int, wsize: int,
                 rbase: int = 0, rhi: int = -1):
    """Fused V-cycle residual y = b - A@x (x given as window pieces)."""
    hlo, own, hhi = pieces
    ext().dia_residual(dm.dvals, dm.offs, hlo.contiguous(), own.contiguous(),
                       hhi.contiguous(), b, y, dm.W, dm.m, int(col_lo),
                       dm.row0, int(wsize), int(rbase), int(rhi))


def dia_jacobi(dm: DiaMirror, pieces, xloc, b, dinv, omega, xout,
               col_lo: int, wsize: int, rbase: int = 0, rhi: int = -1):
    hlo, own, hhi = pieces
    ext().dia_jacobi(dm.dvals, dm.offs, hlo.contiguous(), own.contiguous(),
                     hhi.contiguous(), xloc, b, dinv, xout, dm.W, dm.m,
                     int(col_lo), dm.row0, int(wsize), float(omega),
                     int(rbase), int(rhi))


# -- BSR (MFMA) fast path -----------------------------------------------------
class BsrMirror:
    """16x16 dense-block mirror of a CSR slab for the MFMA SpMM path
    (profiles/MFMA_r02.md): bvals holds row-major 16x16 blocks, bcol the
    GLOBAL block-column ids.  Local block rows cover the slab rows 0..m."""

    __slots__ = ("bptr", "bcol", "bvals", "nbrows", "m", "fill")

    def __init__(self, bptr, bcol, bvals, nbrows, m, fill):
        self.bptr = bptr
        self.bcol = bcol
        self.bvals = bvals
        self.nbrows = nbrows
        self.m = m
        self.fill = fill


def build_bsr(A, min_fill=0.05):
    """Build the 16x16 BSR mirror of a LocalCSR (fp32/fp64 only), or None
    when the block fill ratio / memory budget make MFMA unprofitable."""
    m = A.nrows
    if m == 0 or A.nnz == 0 or A.values.dtype not in (torch.float32,
                                                      torch.float64):
        return None
    dev = A.device
    counts = A.indptr[1:] - A.indptr[:-1]
    lrows = torch.repeat_interleave(
        torch.arange(m, dtype=torch.int64, device=dev), counts)
    brow = lrows >> 4
    bcol_nnz = A.indices.long() >> 4
    nbc = (A.ncols + 15) // 16
    key = brow * nbc + bcol_nnz
    ukey = torch.unique(key)
    nblocks = int(ukey.numel())
    fill = A.nnz / (256.0 * nblocks)
    if fill < min_fill:
        return None
    need = nblocks * 256 * A.values.element_size()
    free, _t = torch.cuda.mem_get_info(dev) if A.values.is_cuda else (1 << 62, 0)
    if need > 0.25 * free:
        return None
    nbrows = (m + 15) // 16
    blk = torch.searchsorted(ukey, key)
    bvals = torch.zeros(nblocks * 256, dtype=A.values.dtype, device=dev)
    bvals[blk * 256 + (lrows & 15) * 16 + (A.indices.long() & 15)] = A.values
    bptr = torch.zeros(nbrows + 1, dtype=torch.int64, device=dev)
    bptr[1:] = torch.cumsum(
        torch.bincount(ukey // nbc, minlength=nbrows), dim=0)
    bcol = (ukey - (ukey // nbc) * nbc).to(torch.int32)
    return BsrMirror(bptr, bcol, bvals, nbrows, m, fill)


def bsr_spmm(bm: BsrMirror, Bw, C, col_lo: int):
    """C[m, k] = BSR @ Bw (gathered window rows, global cols - col_lo)."""
    ext().bsr_spmm(bm.bptr, bm.bcol, bm.bvals, Bw, C, int(col_lo))


def bsr_profitable(bm: BsrMirror, k: int) -> bool:
    """Measured win region (profiles/MFMA_r02.md): at k<=32 MFMA wins from
    ~6% fill; by k=64 the lane-tiled kernel catches up below ~20% fill."""
    return bm is not None and (k <= 32 and bm.fill >= 0.06
                               or bm.fill >= 0.2)


# -- ELL fast path ------------------------------------------------------------
class EllMirror:
    """Column-major padded-ELL copy of a row-uniform CSR slab (fast SpMV).
    rmin/rmax are per-row global column windows (for the ws>1
    interior/boundary overlap split)."""

    __slots__ = ("eidx", "evals", "W", "m", "rmin", "rmax")

    def __init__(self, eidx, evals, W, m, rmin=None, rmax=None):
        self.eidx = eidx
        self.evals = evals
        self.W = W
        self.m = m
        self.rmin = rmin
        self.rmax = rmax


def build_ell(A):
    """Build the ELL mirror of a LocalCSR, or None when unprofitable
    (W too large or padding blowup > 1.6x the CSR bytes)."""
    m = A.nrows
    if m == 0 or A.nnz == 0:
        return None
    counts = A.indptr[1:] - A.indptr[:-1]
    W = int(counts.max().item())
    if W == 0 or W > 48:
        return None
    mp = (m + 1) // 2 * 2
    if W * mp > 1.6 * A.nnz + 4096:
        return None
    need = W * mp * (A.values.element_size() + A.indices.element_size())
    free, _total = torch.cuda.mem_get_info(A.values.device)
    if need > 0.5 * free:
        # mirror would not fit comfortably — stay on the CSR kernels
        return None
    eidx = torch.empty(W * mp, dtype=A.indices.dtype, device=A.device)
    evals = torch.empty(W * mp, dtype=A.values.dtype, device=A.device)
    pad_idx = int(A.indices[0].item())
    ext().build_ell(A.indptr, A.indices, A.values, eidx, evals, W, pad_idx)
    # per-row global column windows (empty rows: [n, -1) -> never interior)
    rows = torch.repeat_interleave(
        torch.arange(m, dtype=torch.int64, device=A.device), counts)
    idx = A.indices.long()
    rmin = torch.full((m,), A.ncols, dtype=torch.int64, device=A.device)
    rmax = torch.full((m,), -1, dtype=torch.int64, device=A.device)
    rmin.scatter_reduce_(0, rows, idx, "amin")
    rmax.scatter_reduce_(0, rows, idx, "amax")
    return EllMirror(eidx, evals, W, m, rmin, rmax)


def ell_interior(ell: EllMirror, own_a: int, own_b: int):
    """Even row bounds [a, b) of the longest contiguous run of rows whose
    whole column window lies in the own x piece [own_a, own_b) — the ELL
    analog of the DIA interior split.  Returns (0, 0) when the interior
    is too small to be worth a split kernel pair."""
    mask = (ell.rmin >= own_a) & (ell.rmax < own_b)
    pad = torch.zeros(1, dtype=torch.int8, device=mask.device)
    d = torch.diff(torch.cat([pad, mask.to(torch.int8), pad]))
    starts = (d == 1).nonzero(as_tuple=False).flatten()
    ends = (d == -1).nonzero(as_tuple=False).flatten()
    if starts.numel() == 0:
        return 0, 0
    lens = ends - starts
    j = int(torch.argmax(lens).item())
    a, b = int(starts[j].item()), int(ends[j].item())
    a = (a + 1) // 2 * 2
    b = max(a, b // 2 * 2)
    if b - a < max(1024, ell.m // 4):
        return 0, 0
    return a, b


def ell_spmv(ell: EllMirror, pieces, y, col_lo: int,
             rbase: int = 0, rhi: int = -1):
    hlo, own, hhi = pieces
    ext().ell_spmv(ell.eidx, ell.evals, hlo.contiguous(), own.contiguous(),
                   hhi.contiguous(), y, ell.W, ell.m, int(col_lo),
                   int(rbase), int(rhi))


def ell_jacobi(ell: EllMirror, pieces, xloc, b, dinv, omega, xout, col_lo: int):
    """Fused weighted-Jacobi sweep: xout = x + omega*dinv*(b - A x)."""
    hlo, own, hhi = pieces
    ext().ell_jacobi(ell.eidx, ell.evals, hlo.contiguous(), own.contiguous(),
                     hhi.contiguous(), xloc, b, dinv, xout, ell.W, ell.m,
                     int(col_lo), float(omega))


def ell_spmv_dot(ell: EllMirror, pieces, y, p, col_lo: int,
                 rbase: int = 0, rhi: int = -1):
    hlo, own, hhi = pieces
    mp = ell.evals.numel() // ell.W
    hi = mp if rhi < 0 else rhi
    if hi <= rbase:
        return torch.zeros((), dtype=ell.evals.dtype, device=ell.evals.device)
    nblocks = ((hi - rbase) // 2 + 255) // 256 + 1
    partial = torch.zeros(nblocks, dtype=ell.evals.dtype, device=ell.evals.device)
    ext().ell_spmv_dot(ell.eidx, ell.evals, hlo.contiguous(), own.contiguous(),
                       hhi.contiguous(), y, p, partial, ell.W, ell.m,
                       int(col_lo), int(rbase), int(hi))
    return partial.sum()
