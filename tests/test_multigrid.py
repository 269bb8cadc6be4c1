"""Unit tests for sparse.multigrid.ReplicatedCoarseCycle (the ws>1
coarse-tail latency plan; end-to-end coverage lives in the gmg/amg
example batteries)."""
import numpy as np
import pytest
import scipy.sparse as sps
import torch

from sparse import csr_array, darray
from sparse.multigrid import ReplicatedCoarseCycle, find_replication_cut


class _Lvl:
    pass


def _two_level(n):
    """Tiny 2-level hierarchy on a 1-D Poisson chain with injection-style
    transfers; returns (levels, coarse_inv_t, oracle scipy pieces)."""
    A = sps.diags([-np.ones(n - 1), 2 * np.ones(n), -np.ones(n - 1)],
                  [-1, 0, 1], format="csr")
    nc = n // 2
    P = sps.csr_matrix((np.ones(nc), (2 * np.arange(nc), np.arange(nc))),
                       shape=(n, nc))
    R = P.T.tocsr()
    Ac = (R @ A @ P).tocsr()
    lv0 = _Lvl()
    lv0.A = csr_array(A)
    lv0.dinv = darray.asdistarray(1.0 / A.diagonal())
    lv0.omega = 0.5
    lv0.Rdown = csr_array(R)
    lv0.Pdown = csr_array(P)
    lv1 = _Lvl()
    lv1.A = csr_array(Ac)
    coarse_inv = torch.as_tensor(np.linalg.pinv(Ac.toarray()))
    return [lv0, lv1], coarse_inv, (A, P, R, Ac)


def test_replicated_cycle_matches_explicit_vcycle():
    n = 64
    levels, cinv, (A, P, R, Ac) = _two_level(n)
    cyc = ReplicatedCoarseCycle(levels, cinv, smooth_iters=2)
    b = torch.as_tensor(np.random.default_rng(5).random(n))
    got = cyc.apply(b).numpy()

    # explicit oracle of the same V(1,2) cycle
    dinv = 1.0 / A.diagonal()
    om = 0.5
    x = om * dinv * b.numpy()
    x = x + om * dinv * (b.numpy() - A @ x)          # pre-smooth (iters-1=1)
    r = b.numpy() - A @ x
    xc = np.linalg.pinv(Ac.toarray()) @ (R @ r)      # coarse solve
    x = x + P @ xc
    for _ in range(2):                               # post-smooth (iters=2)
        x = x + om * dinv * (b.numpy() - A @ x)
    assert np.allclose(got, x, rtol=1e-12, atol=1e-12)


def test_replicated_cycle_is_spd_preconditioner_quality():
    """Used as M in CG it must converge fast on the chain Poisson."""
    import scipy.sparse.linalg as spla

    n = 128
    levels, cinv, (A, *_rest) = _two_level(n)
    cyc = ReplicatedCoarseCycle(levels, cinv, smooth_iters=2)
    b = np.random.default_rng(6).random(n)
    it = [0]
    x, info = spla.cg(
        A, b, rtol=1e-10, maxiter=200,
        M=spla.LinearOperator((n, n),
                              matvec=lambda r: cyc.apply(
                                  torch.as_tensor(r)).numpy()),
        callback=lambda xk: it.__setitem__(0, it[0] + 1))
    assert info == 0
    assert it[0] < 100  # unpreconditioned CG needs ~n=128 iterations here


def test_find_replication_cut():
    levels, _, _ = _two_level(40)
    assert find_replication_cut(levels, 0) == 2  # nothing at/below 0 rows
    assert find_replication_cut(levels, 25) == 1  # only the coarse level
    assert find_replication_cut(levels, 1000) == 0  # everything
    # dict-style levels (amg.py shape)
    dl = [{"A": levels[0].A}, {"A": levels[1].A}]
    assert find_replication_cut(dl, 25) == 1
