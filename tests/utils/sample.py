"""Seeded random matrices (role of reference utils/sample.py:25-44)."""
import numpy as np
import scipy.sparse as sps


def sample_csr(m, n, density=0.3, seed=0, dtype=np.float64):
    rng = np.random.default_rng(seed)
    s = sps.random(m, n, density=density, random_state=rng, format="csr")
    if np.issubdtype(np.dtype(dtype), np.complexfloating):
        s = s.astype(dtype)
        s.data += 1j * rng.random(len(s.data)).astype(s.data.real.dtype)
    else:
        s = s.astype(dtype)
    s.sort_indices()
    return s


def sample_dense(shape, seed=0, dtype=np.float64):
    rng = np.random.default_rng(seed)
    d = rng.random(shape)
    if np.issubdtype(np.dtype(dtype), np.complexfloating):
        return (d + 1j * rng.random(shape)).astype(dtype)
    return d.astype(dtype)


def spd_csr(n, density=0.3, seed=0, dtype=np.float64):
    """SPD matrix A + A^T + n*I (reference test_cg_solve.py:23-36)."""
    s = sample_csr(n, n, density, seed, np.float64)
    a = (s + s.T + n * sps.eye(n)).tocsr().astype(dtype)
    if np.issubdtype(np.dtype(dtype), np.complexfloating):
        a = (a + a.conj().T).tocsr() / 2
    a.sort_indices()
    return a
