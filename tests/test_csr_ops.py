"""CSR products: SpMV / SpMM / SpGEMM / SDDMM / rSpMM vs scipy oracle.

Coverage parity: reference test_csr_dot.py, test_csr_spmm.py,
test_csr_sddmm.py, test_csr_spgemm.py.
"""
import numpy as np
import pytest

import sparse
from sparse import csr_array

from utils.common import types
from utils.sample import sample_csr, sample_dense


def tol(dt):
    if np.dtype(dt) in (np.float32, np.complex64):
        return dict(rtol=2e-4, atol=2e-5)
    return dict(rtol=1e-10, atol=1e-12)


@pytest.mark.parametrize("mat_type", types)
@pytest.mark.parametrize("vec_type", types)
def test_spmv(mat_type, vec_type):
    s = sample_csr(17, 23, 0.4, seed=1, dtype=mat_type)
    x = sample_dense(23, seed=2, dtype=vec_type)
    A = csr_array(s)
    y = A @ x
    expected = s @ x
    assert np.allclose(np.asarray(y), expected, **tol(np.promote_types(mat_type, vec_type)))


@pytest.mark.parametrize("dt", types)
def test_spmv_rectangular_tall(dt):
    s = sample_csr(40, 7, 0.5, seed=3, dtype=dt)
    x = sample_dense(7, seed=4, dtype=dt)
    assert np.allclose(np.asarray(csr_array(s) @ x), s @ x, **tol(dt))


@pytest.mark.parametrize("dt", types)
def test_spmm(dt):
    s = sample_csr(13, 19, 0.4, seed=5, dtype=dt)
    B = sample_dense((19, 6), seed=6, dtype=dt)
    C = csr_array(s) @ B
    assert np.allclose(np.asarray(C), s @ B, **tol(dt))


@pytest.mark.parametrize("dt", types)
def test_rspmm(dt):
    s = sample_csr(11, 9, 0.5, seed=7, dtype=dt)
    A = sample_dense((4, 11), seed=8, dtype=dt)
    C = A @ csr_array(s)
    assert np.allclose(np.asarray(C), A @ s, **tol(dt))


@pytest.mark.parametrize("dt", types)
def test_spgemm_csr_csr(dt):
    a = sample_csr(14, 18, 0.3, seed=9, dtype=dt)
    b = sample_csr(18, 12, 0.3, seed=10, dtype=dt)
    C = csr_array(a) @ csr_array(b)
    assert np.allclose(np.asarray(C.todense()), (a @ b).toarray(), **tol(dt))


@pytest.mark.parametrize("dt", types)
def test_spgemm_csr_csc(dt):
    a = sample_csr(10, 15, 0.35, seed=11, dtype=dt)
    b = sample_csr(15, 9, 0.35, seed=12, dtype=dt)
    C = csr_array(a) @ sparse.csc_array(b.tocsc())
    assert np.allclose(np.asarray(C.todense()), (a @ b).toarray(), **tol(dt))


@pytest.mark.parametrize("dt", [np.float32, np.float64])
def test_sddmm(dt):
    s = sample_csr(9, 11, 0.5, seed=13, dtype=dt)
    C = sample_dense((9, 5), seed=14, dtype=dt)
    D = sample_dense((5, 11), seed=15, dtype=dt)
    out = csr_array(s).sddmm(C, D)
    expected = s.multiply(C @ D).toarray()
    assert np.allclose(np.asarray(out.todense()), expected, **tol(dt))


def test_spmv_out_param():
    s = sample_csr(12, 12, 0.4, seed=16)
    A = csr_array(s)
    x = sample_dense(12, seed=17)
    out = sparse.darray.zeros((12,))
    r = A.dot(sparse.asdistarray(x), out=out)
    assert r is out
    assert np.allclose(np.asarray(out), s @ x)


def test_matvec_transpose():
    s = sample_csr(8, 13, 0.5, seed=18)
    x = sample_dense(8, seed=19)
    assert np.allclose(np.asarray(csr_array(s).T @ x), s.T @ x)


def test_dot_vector_matmul_operator():
    s = sample_csr(6, 6, 0.6, seed=20)
    x = sample_dense(6, seed=21)
    # x @ A (vector from the left)
    assert np.allclose(np.asarray(x @ csr_array(s)), x @ s)


def test_spgemm_dense_products_cpu():
    """Dense-product multiply sanity on the CPU path (the GPU checked-bin
    and ESC-batching variants live in test_gpu_kernels)."""
    import scipy.sparse as sps

    from sparse import csr_array

    a = sps.random(60, 50, 0.4, random_state=45).tocsr()
    b = sps.random(50, 2000, 0.6, random_state=46).tocsr()
    C = csr_array(a) @ csr_array(b)
    ref = (a @ b).tocsr()
    assert C.nnz == ref.nnz
    assert np.allclose(np.asarray(C.todense()), ref.toarray(), rtol=1e-10)


def test_indexing():
    """A[i], A[i:j], A[i, j] (superset: the reference supports no
    indexing)."""
    s = sample_csr(12, 9, 0.4, seed=55)
    A = csr_array(s)
    assert np.allclose(np.asarray(A[3].todense()), s[[3]].toarray())
    assert np.allclose(np.asarray(A[2:9].todense()), s[2:9].toarray())
    assert np.allclose(np.asarray(A[-1].todense()), s[[-1]].toarray())
    assert np.isclose(A[3, 4], s[3, 4])
    assert A[5:5].shape == (0, 9)
    with pytest.raises(IndexError):
        A[99]
