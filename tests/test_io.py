"""mmread/mmwrite (coverage parity: reference test_io.py)."""
import numpy as np
import scipy.io as sio

import sparse.io as sio_ours

from utils.common import test_mtx_files


def test_mmread_matches_scipy():
    for path in test_mtx_files:
        ours = sio_ours.mmread(path)
        ref = sio.mmread(path)
        assert ours.shape == ref.shape, path
        assert np.allclose(np.asarray(ours.todense()), ref.toarray()), path


def test_mmread_tocsr():
    path = test_mtx_files[0]
    A = sio_ours.mmread(path).tocsr()
    ref = sio.mmread(path).tocsr()
    assert np.allclose(np.asarray(A.todense()), ref.toarray())


def test_mmwrite_roundtrip(tmp_path):
    import sparse

    A = sparse.random(12, 9, 0.3, random_state=1, format="csr")
    p = str(tmp_path / "out.mtx")
    sio_ours.mmwrite(p, A)
    back = sio_ours.mmread(p)
    assert np.allclose(np.asarray(back.todense()), np.asarray(A.todense()))


def test_mmread_integer_field(tmp_path):
    p = tmp_path / "i.mtx"
    p.write_text("%%MatrixMarket matrix coordinate integer general\n"
                 "3 3 3\n1 1 5\n2 3 -2\n3 2 7\n")
    A = sio_ours.mmread(str(p))
    ref = np.zeros((3, 3))
    ref[0, 0], ref[1, 2], ref[2, 1] = 5, -2, 7
    assert np.allclose(np.asarray(A.todense()), ref)


def test_mmread_skew_symmetric(tmp_path):
    import scipy.io as sio

    p = tmp_path / "s.mtx"
    p.write_text("%%MatrixMarket matrix coordinate real skew-symmetric\n"
                 "3 3 2\n2 1 1.5\n3 1 -2.0\n")
    A = sio_ours.mmread(str(p))
    assert np.allclose(np.asarray(A.todense()), sio.mmread(str(p)).toarray())


def test_mmread_hermitian(tmp_path):
    """Hermitian coordinate files: the stored lower triangle expands with
    CONJUGATED mirror entries (reference mtx_to_coo.cc symmetry handling)."""
    import scipy.io as spio
    import scipy.sparse as sps

    rng = np.random.default_rng(44)
    n = 12
    low = sps.random(n, n, 0.3, random_state=45).astype(np.complex128)
    low.data = low.data + 1j * rng.random(low.nnz)
    low = sps.tril(low)
    # make diagonal real so the matrix is a valid hermitian
    full = low + low.conj().T
    full.setdiag(full.diagonal().real)
    p = str(tmp_path / "herm.mtx")
    spio.mmwrite(p, full.tocoo(), symmetry="hermitian")
    ours = sio_ours.mmread(p).tocsr().to_scipy_sparse_csr()
    ref = spio.mmread(p).tocsr()
    ref.sort_indices()
    assert np.allclose(ours.toarray(), ref.toarray())
    assert np.allclose(ours.toarray(), ours.toarray().conj().T)
