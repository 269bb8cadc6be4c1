"""Module constructors (coverage parity: reference test_module.py)."""
import numpy as np
import pytest
import scipy.sparse as sps

import sparse
from sparse import csr_array


def test_eye():
    for m, n, k in [(5, None, 0), (5, 7, 0), (6, 4, -1), (5, 5, 2)]:
        ours = sparse.eye(m, n, k=k)
        ref = sps.eye(m, n if n is not None else m, k=k)
        assert np.allclose(np.asarray(ours.todense()), ref.toarray()), (m, n, k)


def test_identity():
    assert np.allclose(np.asarray(sparse.identity(6).todense()), np.eye(6))


def test_diags_variants():
    cases = [
        ([[1, 2, 3, 4]], [0]),
        ([[1, 2, 3], [4, 5, 6, 7], [8, 9]], [-1, 0, 2]),
        ([np.ones(5), 2 * np.ones(4)], [0, -1]),
    ]
    for diags, offs in cases:
        ours = sparse.diags(diags, offs)
        ref = sps.diags(diags, offs)
        assert np.allclose(np.asarray(ours.todense()), ref.toarray()), (diags, offs)


def test_diags_scalar_broadcast():
    ours = sparse.diags([2.0], [1], shape=(5, 5))
    ref = sps.diags([2.0], [1], shape=(5, 5))
    assert np.allclose(np.asarray(ours.todense()), ref.toarray())


def test_diags_shape():
    ours = sparse.diags([np.arange(1, 6)], [1], shape=(6, 6))
    ref = sps.diags([np.arange(1, 6)], [1], shape=(6, 6))
    assert np.allclose(np.asarray(ours.todense()), ref.toarray())


def test_spdiags():
    data = np.array([[1, 2, 3, 4.0], [5, 6, 7, 8.0]])
    ours = sparse.spdiags(data, [0, 1], 4, 4)
    ref = sps.spdiags(data, [0, 1], 4, 4)
    assert np.allclose(np.asarray(ours.todense()), ref.toarray())


def test_kron():
    a = sps.random(4, 3, 0.5, random_state=1).tocsr()
    b = sps.random(2, 5, 0.6, random_state=2).tocsr()
    ours = sparse.kron(csr_array(a), csr_array(b), format="csr")
    ref = sps.kron(a, b)
    assert np.allclose(np.asarray(ours.todense()), ref.toarray())


def test_random():
    r = sparse.random(30, 20, density=0.1, random_state=3, format="csr")
    assert r.shape == (30, 20)
    assert r.nnz == int(round(0.1 * 600))
    d = np.asarray(r.todense())
    assert (d != 0).sum() == r.nnz


def test_random_dedup_sampler_high_density():
    """Huge-matrix sampler must return exactly nnz distinct indices even
    when collisions exceed the first-draw margin (high density)."""
    from sparse.module import _sample_flat_dedup

    rng = np.random.default_rng(7)
    mn, nnz = 1000, 950  # 95% density: one 5%-margin draw WILL collide short
    flat = _sample_flat_dedup(rng, mn, nnz)
    assert flat.shape[0] == nnz
    assert np.unique(flat).shape[0] == nnz
    assert flat.min() >= 0 and flat.max() < mn


def test_fused_norm_wrappers_empty_slab():
    """axpby_norm2 / cg_xr_norm2 must contribute an exact 0 for an empty
    local slab (uninitialized partial buffer corrupted CG)."""
    import torch
    from sparse import kernels

    e = torch.zeros(0, dtype=torch.float64)
    z = kernels.axpby_norm2(e, e, torch.tensor(1.0), torch.tensor(1.0),
                            True, False)
    assert float(z) == 0.0
    z2 = kernels.cg_xr_norm2(e, e, e, e, torch.tensor(1.0), torch.tensor(1.0))
    assert float(z2) == 0.0


def test_issparse_predicates():
    A = sparse.eye(3)
    assert sparse.issparse(A)
    assert sparse.isspmatrix_csr(A)
    assert not sparse.isspmatrix_coo(A)
    assert sparse.isspmatrix_coo(A.tocoo())
    assert sparse.isspmatrix_csc(A.tocsc())
    assert sparse.isspmatrix_dia(A.tocoo().todia())
    assert not sparse.issparse(np.eye(3))


def test_stack_and_triangles():
    """hstack/vstack/bmat/block_diag/tril/triu (superset: the reference's
    namespace clone does not provide these)."""
    import scipy.sparse as sps

    import sparse

    a = sps.random(6, 8, 0.4, random_state=1, format="csr")
    b = sps.random(6, 5, 0.4, random_state=2, format="csr")
    c = sps.random(4, 8, 0.4, random_state=3, format="csr")
    H = sparse.hstack([sparse.csr_array(a), sparse.csr_array(b)])
    assert np.allclose(np.asarray(H.todense()),
                       sps.hstack([a, b]).toarray())
    V = sparse.vstack([sparse.csr_array(a), sparse.csr_array(c)])
    assert np.allclose(np.asarray(V.todense()),
                       sps.vstack([a, c]).toarray())
    B = sparse.bmat([[sparse.csr_array(a), None],
                     [None, sparse.csr_array(b)]])
    assert np.allclose(np.asarray(B.todense()),
                       sps.bmat([[a, None], [None, b]]).toarray())
    D = sparse.block_diag([sparse.csr_array(a), sparse.csr_array(b)])
    assert np.allclose(np.asarray(D.todense()),
                       sps.block_diag([a, b]).toarray())
    sq = sps.random(7, 7, 0.5, random_state=4, format="csr")
    for k in (-2, 0, 1):
        assert np.allclose(np.asarray(sparse.tril(sparse.csr_array(sq), k=k).todense()),
                           sps.tril(sq, k=k).toarray())
        assert np.allclose(np.asarray(sparse.triu(sparse.csr_array(sq), k=k).todense()),
                           sps.triu(sq, k=k).toarray())


def test_bounds_checks_env(monkeypatch):
    """SPARSE_BOUNDS_CHECKS=1 validates structure at construction
    (reference Legion_BOUNDS_CHECKS parity)."""
    import scipy.sparse as sps

    import sparse

    monkeypatch.setenv("SPARSE_BOUNDS_CHECKS", "1")
    s = sps.random(10, 10, 0.3, random_state=1, format="csr")
    A = sparse.csr_array(s)  # valid structure passes
    assert A.nnz == s.nnz
    bad_indices = np.array([0, 99], dtype=np.int64)  # col 99 out of range
    bad_indptr = np.array([0, 1, 2, 2, 2, 2, 2, 2, 2, 2, 2], dtype=np.int64)
    with pytest.raises((ValueError, AssertionError, RuntimeError)):
        sparse.csr_array((np.ones(2), bad_indices, bad_indptr), shape=(10, 10))


def test_abs_dunder():
    s = sps.random(15, 15, 0.3, random_state=8, format="csr") - 0.5 * sps.eye(15)
    s = s.tocsr()
    A = csr_array(s)
    assert np.allclose(np.asarray(abs(A).todense()), abs(s).toarray())


def test_norm_npz_eliminate():
    """linalg.norm, save_npz/load_npz round-trip, eliminate_zeros and the
    sorted-indices compat no-ops (scipy-API supersets)."""
    import os
    import tempfile

    import sparse
    import sparse.linalg as sl

    s = sps.random(20, 30, 0.3, random_state=1, format="csr")
    A = sparse.csr_array(s)
    assert np.isclose(sl.norm(A), sps.linalg.norm(s))
    assert np.isclose(sl.norm(A, 1), sps.linalg.norm(s, 1))
    assert np.isclose(sl.norm(A, np.inf), sps.linalg.norm(s, np.inf))
    with tempfile.TemporaryDirectory() as d:
        p = os.path.join(d, "x.npz")
        sparse.io.save_npz(p, A)
        B = sparse.io.load_npz(p)
        assert np.allclose(np.asarray(B.todense()), s.toarray())
    z = s.copy()
    z.data[::3] = 0.0
    Az = sparse.csr_array(z)
    Az.eliminate_zeros()
    zc = z.copy()
    zc.eliminate_zeros()
    assert Az.nnz == zc.nnz
    assert np.allclose(np.asarray(Az.todense()), zc.toarray())
    assert A.has_sorted_indices and A.sort_indices() is None
    assert A.sum_duplicates() is None
