"""Gallery builders vs scipy constructions."""
import numpy as np
import scipy.sparse as sps

from sparse import gallery


def _poisson2d_ref(nx, ny):
    T = sps.diags([-1, 2, -1], [-1, 0, 1], (nx, nx))
    S = sps.diags([-1, 2, -1], [-1, 0, 1], (ny, ny))
    return (sps.kron(sps.eye(ny), T) + sps.kron(S, sps.eye(nx))).tocsr()


def test_poisson2d():
    A = gallery.poisson2d(7, 5)
    ref = _poisson2d_ref(7, 5)
    assert np.allclose(np.asarray(A.todense()), ref.toarray())


def test_poisson3d():
    nx = 4
    I = sps.eye(nx)
    T = sps.diags([-1, 2, -1], [-1, 0, 1], (nx, nx))
    ref = (sps.kron(sps.kron(I, I), T) + sps.kron(sps.kron(I, T), I)
           + sps.kron(sps.kron(T, I), I)).tocsr()
    A = gallery.poisson3d(nx)
    assert np.allclose(np.asarray(A.todense()), ref.toarray())


def _p1d(nc):
    nf = 2 * nc + 1
    P = sps.lil_matrix((nf, nc))
    for I in range(nc):
        P[2 * I + 1, I] = 1.0
        P[2 * I, I] = 0.5
        if 2 * I + 2 < nf:
            P[2 * I + 2, I] = 0.5
    return P.tocsr()


def test_interpolation2d_sorted_and_correct():
    nx = 9
    ref = sps.kron(_p1d((nx - 1) // 2), _p1d((nx - 1) // 2)).tocsr()
    P = gallery.interpolation2d(nx)
    assert np.allclose(np.asarray(P.todense()), ref.toarray())
    # sorted-indices invariant (required by add/mult kernels)
    ip, ix = P.indptr, P.indices
    for r in range(P.shape[0]):
        seg = ix[ip[r]: ip[r + 1]]
        assert np.all(np.diff(seg) > 0), r


def test_interpolation3d():
    nx = 5
    P1 = _p1d((nx - 1) // 2)
    ref = sps.kron(sps.kron(P1, P1), P1).tocsr()
    P = gallery.interpolation3d(nx)
    assert np.allclose(np.asarray(P.todense()), ref.toarray())
    ip, ix = P.indptr, P.indices
    for r in range(P.shape[0]):
        seg = ix[ip[r]: ip[r + 1]]
        assert np.all(np.diff(seg) > 0), r


def test_banded():
    A = gallery.banded(20, ndiags=5)
    ref = sps.diags([np.ones(20)] * 5, list(range(-2, 3)), (20, 20)).tocsr()
    ref = ref + 4 * sps.diags([np.ones(20)], [0])  # center weight = ndiags
    assert np.allclose(np.asarray(A.todense()), ref.toarray())


def test_spmv_domain_part_flag():
    """dot(..., spmv_domain_part=True) matches plain dot (reference
    csr.py:863-968 semantics; here both map to the row-split path)."""
    from sparse import csr_array

    s = sps.random(30, 30, 0.3, random_state=1, format="csr")
    x = np.random.default_rng(2).random(30)
    A = csr_array(s)
    a = np.asarray(A.dot(x))
    b = np.asarray(A.dot(x, spmv_domain_part=True))
    assert np.allclose(a, b)
    assert np.allclose(a, s @ x)
