"""Format conversions + misc (coverage parity: reference
test_csr_conversion.py, test_csr_misc.py, test_coo.py, test_csc.py,
test_dia.py)."""
import numpy as np
import pytest
import scipy.sparse as sps

import sparse
from sparse import coo_array, csc_array, csr_array, dia_array

from utils.common import types
from utils.sample import sample_csr, sample_dense


@pytest.mark.parametrize("dt", types)
def test_csr_roundtrips(dt):
    s = sample_csr(14, 10, 0.35, seed=1, dtype=dt)
    A = csr_array(s)
    dense = s.toarray()
    assert np.allclose(np.asarray(A.todense()), dense)
    assert np.allclose(np.asarray(A.tocoo().todense()), dense)
    assert np.allclose(np.asarray(A.tocsc().todense()), dense)
    assert np.allclose(np.asarray(A.tocsc().tocsr().todense()), dense)
    assert np.allclose(np.asarray(A.tocoo().tocsr().todense()), dense)


@pytest.mark.parametrize("dt", types)
def test_coo_construct_and_convert(dt):
    s = sample_csr(12, 16, 0.3, seed=2, dtype=dt).tocoo()
    C = coo_array((s.data, (s.row, s.col)), shape=s.shape)
    assert C.nnz == s.nnz
    assert np.allclose(np.asarray(C.todense()), s.toarray())
    assert np.allclose(np.asarray(C.tocsr().todense()), s.toarray())
    assert np.allclose(np.asarray(C.tocsc().todense()), s.toarray())
    assert np.allclose(np.asarray(C.T.todense()), s.T.toarray())


def test_coo_duplicates_summed():
    i = np.array([0, 0, 1, 2, 0])
    j = np.array([1, 1, 2, 0, 1])
    v = np.array([1.0, 2.0, 3.0, 4.0, 5.0])
    C = coo_array((v, (i, j)), shape=(3, 3))
    s = sps.coo_matrix((v, (i, j)), shape=(3, 3))
    assert np.allclose(np.asarray(C.tocsr().todense()), s.toarray())


@pytest.mark.parametrize("dt", types)
def test_csc_ops(dt):
    s = sample_csr(11, 13, 0.4, seed=3, dtype=dt)
    A = csc_array(s.tocsc())
    x = sample_dense(13, seed=4, dtype=dt)
    assert np.allclose(np.asarray(A @ x), s @ x,
                       rtol=1e-4 if np.dtype(dt) in (np.float32, np.complex64) else 1e-10)
    B = sample_dense((13, 3), seed=5, dtype=dt)
    assert np.allclose(np.asarray(A @ B), s @ B,
                       rtol=1e-4 if np.dtype(dt) in (np.float32, np.complex64) else 1e-10)
    assert np.allclose(np.asarray(A.T.todense()), s.T.toarray())


def test_csc_transpose_view_shares_no_copy():
    s = sample_csr(9, 9, 0.5, seed=6)
    A = csr_array(s)
    At = A.T
    assert At.shape == (9, 9)
    assert np.allclose(np.asarray(At.todense()), s.T.toarray())
    # view: modifying A.data reflects in At
    assert At._values is A._values


@pytest.mark.parametrize("dt", [np.float64, np.complex128])
def test_dia(dt):
    d0 = np.arange(1, 7).astype(dt)
    d1 = np.arange(10, 15).astype(dt)
    s = sps.diags([d0, d1], [0, 1], shape=(6, 6)).todia()
    A = dia_array((s.data, s.offsets), shape=(6, 6))
    assert np.allclose(np.asarray(A.todense()), s.toarray())
    assert np.allclose(np.asarray(A.T.todense()), s.T.toarray())
    assert np.allclose(np.asarray(A.tocsr().todense()), s.toarray())
    assert np.allclose(np.asarray(A.tocsc().todense()), s.toarray())
    assert np.allclose(np.asarray(A.tocoo().todense()), s.toarray())
    assert np.allclose(np.asarray(A.diagonal()), s.diagonal())


def test_diagonal_offsets():
    s = sample_csr(10, 10, 0.5, seed=7)
    A = csr_array(s)
    assert np.allclose(np.asarray(A.diagonal()), s.diagonal())
    for k in (-2, 1, 3):
        assert np.allclose(np.asarray(A.diagonal(k=k)), s.diagonal(k=k)), k


def test_sum_mean():
    s = sample_csr(9, 14, 0.4, seed=8)
    A = csr_array(s)
    assert np.isclose(A.sum(), s.sum())
    assert np.allclose(np.asarray(A.sum(axis=1)), np.asarray(s.sum(axis=1)).ravel())
    assert np.allclose(np.asarray(A.sum(axis=0)), np.asarray(s.sum(axis=0)).ravel())
    assert np.isclose(A.mean(), s.mean())


def test_unary_and_power():
    s = sample_csr(8, 8, 0.5, seed=9)
    A = csr_array(s)
    assert np.allclose(np.asarray(A.power(2).todense()), s.power(2).toarray())
    assert np.allclose(np.asarray(A.sqrt().todense()), np.sqrt(s.toarray()))
    assert np.allclose(np.asarray(abs(-A).todense() if hasattr(A, "__abs__") else A.abs().todense()),
                       abs(s.toarray()))


def test_astype_copy_conj():
    s = sample_csr(7, 7, 0.5, seed=10, dtype=np.complex128)
    A = csr_array(s)
    assert A.astype(np.complex64).dtype == np.complex64
    assert np.allclose(np.asarray(A.conj().todense()), s.conj().toarray())
    B = A.copy()
    B.data = B.data * 0
    assert not np.allclose(np.asarray(B.todense()), np.asarray(A.todense())) or s.nnz == 0


def test_balance_noop_single_rank():
    s = sample_csr(20, 20, 0.3, seed=11)
    A = csr_array(s)
    A.balance()
    assert np.allclose(np.asarray(A.todense()), s.toarray())


def test_indptr_indices_data_roundtrip():
    s = sample_csr(10, 12, 0.4, seed=12)
    A = csr_array(s)
    assert np.array_equal(A.indptr, s.indptr.astype(np.int64))
    assert np.array_equal(A.indices, s.indices)
    assert np.allclose(A.data, s.data)
    B = csr_array((A.data, A.indices, A.indptr), shape=A.shape)
    assert np.allclose(np.asarray(B.todense()), s.toarray())


def test_empty_and_shape_ctor():
    A = csr_array((4, 5))
    assert A.nnz == 0 and A.shape == (4, 5)
    assert np.allclose(np.asarray(A.todense()), np.zeros((4, 5)))


def test_jacobi_smooth_cpu():
    import sparse
    from sparse import darray, gallery

    A = gallery.poisson2d(16)
    n = A.shape[0]
    x = darray.random((n,), seed=60)
    b = darray.random((n,), seed=61)
    d = A.diagonal()
    dinv = darray.DistArray.from_local(1.0 / d.local, d.partition, d.shape)
    out = A.jacobi_smooth(x, b, dinv, 0.7)
    r = b - A.dot(x)
    expect = np.asarray(x) + 0.7 * np.asarray(dinv) * np.asarray(r)
    assert np.allclose(np.asarray(out), expect, rtol=1e-12)


def test_make_with_same_nnz_structure():
    s = sample_csr(9, 11, 0.4, seed=70)
    A = csr_array(s)
    newvals = np.arange(1.0, s.nnz + 1.0)
    B = csr_array.make_with_same_nnz_structure(A, newvals)
    ref = s.copy()
    ref.data = newvals
    assert np.allclose(np.asarray(B.todense()), ref.toarray())
    assert B._indices is A._indices  # structure shared


def test_csc_sddmm():
    """CSC SDDMM (reference test_csr_sddmm.py's CSC half, CSC_SDDMM task)."""
    import numpy as np
    import scipy.sparse as sps

    import sparse

    s = sps.random(40, 30, 0.2, random_state=5, format="csc")
    A = sparse.csc_array(s)
    rng = np.random.default_rng(6)
    C = rng.random((40, 4))
    D = rng.random((4, 30))
    out = A.sddmm(C, D)
    assert np.allclose(np.asarray(out.todense()),
                       s.multiply(C @ D).toarray())


def test_dia_truncated_data_width():
    """scipy DIA data may be narrower than max(k,0)+diag_length (trailing
    zeros truncated): tocoo/tocsr/transpose/diagonal must treat missing
    entries as zero (fuzz-found regression)."""
    import scipy.sparse as sps

    from sparse import csr_array

    rng = np.random.default_rng(987654)
    for _ in range(5):
        rng.integers(1, 50, 3)
    m = n = 14
    a = sps.random(m, n, 0.45, random_state=661633, format="csr")
    a.sort_indices()
    di = csr_array(a).tocoo().todia()
    assert np.allclose(np.asarray(di.todense()), a.toarray())
    assert np.allclose(np.asarray(di.tocsr().todense()), a.toarray())
    assert np.allclose(np.asarray(di.T.todense()), a.T.toarray())
    assert np.allclose(np.asarray(di.diagonal()), a.diagonal())
