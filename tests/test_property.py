"""Property-based fuzzing vs the scipy oracle (hypothesis).

Random shapes/densities/dtypes through the core op surface — catches
edge cases (empty rows, duplicate-free randoms, single-column windows)
that the fixed-seed tests miss.
"""
import numpy as np
import scipy.sparse as sps
from hypothesis import given, settings, strategies as st

from sparse import csr_array, csc_array

DT = st.sampled_from([np.float64, np.float32])


def _rand(m, n, density, seed, dt):
    s = sps.random(m, n, density, random_state=seed, format="csr",
                   dtype=np.float64)
    s.sort_indices()
    return s.astype(dt)


@settings(max_examples=25, deadline=None, derandomize=True)
@given(m=st.integers(1, 40), n=st.integers(1, 40),
       density=st.floats(0.0, 0.5), seed=st.integers(0, 10**6), dt=DT)
def test_spmv_matches_scipy(m, n, density, seed, dt):
    s = _rand(m, n, density, seed, dt)
    x = np.random.default_rng(seed).random(n).astype(dt)
    got = np.asarray(csr_array(s) @ x)
    rtol = 1e-10 if dt == np.float64 else 1e-4
    assert np.allclose(got, s @ x, rtol=rtol, atol=1e-6)


@settings(max_examples=20, deadline=None, derandomize=True)
@given(m=st.integers(1, 30), n=st.integers(1, 30), k=st.integers(1, 30),
       d1=st.floats(0.05, 0.5), d2=st.floats(0.05, 0.5),
       seed=st.integers(0, 10**6))
def test_spgemm_matches_scipy(m, n, k, d1, d2, seed):
    a = _rand(m, n, d1, seed, np.float64)
    b = _rand(n, k, d2, seed + 1, np.float64)
    C = csr_array(a) @ csr_array(b)
    ref = (a @ b).tocsr()
    assert C.shape == ref.shape
    assert np.allclose(np.asarray(C.todense()), ref.toarray(),
                       rtol=1e-10, atol=1e-12)


@settings(max_examples=20, deadline=None, derandomize=True)
@given(m=st.integers(1, 30), n=st.integers(1, 30),
       d1=st.floats(0.0, 0.5), d2=st.floats(0.0, 0.5),
       seed=st.integers(0, 10**6))
def test_add_mult_match_scipy(m, n, d1, d2, seed):
    a = _rand(m, n, d1, seed, np.float64)
    b = _rand(m, n, d2, seed + 7, np.float64)
    S = csr_array(a) + csr_array(b)
    assert np.allclose(np.asarray(S.todense()), (a + b).toarray())
    M = csr_array(a).multiply(csr_array(b))
    assert np.allclose(np.asarray(M.todense()), a.multiply(b).toarray())


@settings(max_examples=15, deadline=None, derandomize=True)
@given(m=st.integers(1, 30), n=st.integers(1, 30),
       density=st.floats(0.0, 0.5), seed=st.integers(0, 10**6))
def test_conversion_roundtrips(m, n, density, seed):
    s = _rand(m, n, density, seed, np.float64)
    A = csr_array(s)
    assert np.allclose(np.asarray(A.tocsc().tocsr().todense()), s.toarray())
    assert np.allclose(np.asarray(A.tocoo().tocsr().todense()), s.toarray())
    assert np.allclose(np.asarray(A.T.todense()), s.T.toarray())
    d = np.asarray(A.diagonal())
    assert np.allclose(d, s.diagonal())


@settings(max_examples=15, deadline=None, derandomize=True)
@given(m=st.integers(1, 30), n=st.integers(1, 30), d=st.floats(0.0, 0.5),
       seed=st.integers(0, 10**6), dt=st.sampled_from([np.float64, np.complex128]))
def test_mmread_roundtrip_matches_scipy(m, n, d, seed, dt):
    """scipy mmwrite -> byte-range-parallel mmread (random shapes incl.
    empty matrices, complex fields)."""
    import os
    import tempfile

    import scipy.io as spio

    import sparse

    s = sps.random(m, n, d, random_state=seed, format="coo").astype(dt)
    if dt == np.complex128 and s.nnz:
        s.data = s.data + 1j * np.random.default_rng(seed).random(s.nnz)
    fd, path = tempfile.mkstemp(suffix=".mtx")
    os.close(fd)
    try:
        spio.mmwrite(path, s)
        back = sparse.io.mmread(path).tocsr().to_scipy_sparse_csr()
    finally:
        os.remove(path)
    ref = s.tocsr()
    ref.sort_indices()
    assert back.shape == ref.shape
    assert np.allclose(back.toarray(), ref.toarray())


@settings(max_examples=15, deadline=None, derandomize=True)
@given(m=st.integers(1, 25), n=st.integers(1, 25), k=st.integers(1, 6),
       d=st.floats(0.05, 0.5), seed=st.integers(0, 10**6))
def test_sddmm_matches_oracle(m, n, k, d, seed):
    s = _rand(m, n, d, seed, np.float64)
    rng = np.random.default_rng(seed)
    C = rng.random((m, k))
    D = rng.random((k, n))
    out = csr_array(s).sddmm(C, D)
    coo = s.tocoo()
    exp = s.multiply(sps.coo_matrix(((C @ D)[coo.row, coo.col],
                                     (coo.row, coo.col)), shape=s.shape))
    assert np.allclose(np.asarray(out.todense()), exp.toarray(),
                       rtol=1e-10, atol=1e-12)


@settings(max_examples=12, deadline=None, derandomize=True)
@given(n=st.integers(5, 50), d=st.floats(0.05, 0.4), seed=st.integers(0, 10**6),
       solver=st.sampled_from(["cg", "minres", "gmres", "bicgstab"]))
def test_solvers_converge_on_spd(n, d, seed, solver):
    """Random SPD systems: every solver must reach the oracle solution."""
    from sparse import linalg

    r = sps.random(n, n, d, random_state=seed, format="csr")
    s = (r + r.T + 2 * n * sps.eye(n)).tocsr()
    b = np.random.default_rng(seed).random(n)
    fn = getattr(linalg, solver)
    kw = {"tol": 1e-11, "maxiter": 3000}
    if solver in ("cg", "bicgstab"):
        kw["conv_test_iters"] = 1
    x, info = fn(csr_array(s), b, **kw)
    assert info == 0, solver
    assert np.allclose(s @ np.asarray(x), b, atol=1e-6), solver


@settings(max_examples=12, deadline=None, derandomize=True)
@given(m=st.integers(2, 25), n=st.integers(2, 25), d=st.floats(0.1, 0.6),
       seed=st.integers(0, 10**6))
def test_transpose_roundtrips(m, n, d, seed):
    s = _rand(m, n, d, seed, np.float64)
    A = csr_array(s)
    assert np.allclose(np.asarray(A.T.T.todense()), s.toarray())
    assert np.allclose(np.asarray(A.T.tocsr().todense()), s.T.toarray())
    assert np.allclose(np.asarray(A.tocsc().tocsr().todense()), s.toarray())


@settings(max_examples=20, deadline=None, derandomize=True)
@given(m=st.integers(1, 40), n=st.integers(1, 40), d=st.floats(0.0, 0.6),
       seed=st.integers(0, 10**6))
def test_format_chain_roundtrip(m, n, d, seed):
    """csr -> coo -> dia -> csr preserves the matrix (the dia leg found a
    truncated-data-width bug in round 2)."""
    s = _rand(m, n, d, seed, np.float64)
    A = csr_array(s)
    back = A.tocoo().todia().tocsr()
    assert np.allclose(np.asarray(back.todense()), s.toarray())
