"""solve_ivp vs scipy oracle (coverage for reference integrate.py usage)."""
import numpy as np
import pytest
from scipy.integrate import solve_ivp as sp_solve_ivp

from sparse import integrate


def exp_decay(t, y):
    return y * (-0.5)


@pytest.mark.parametrize("method", ["RK23", "RK45", "DOP853"])
def test_exponential_decay(method):
    y0 = np.linspace(1.0, 2.0, 20)
    res = integrate.solve_ivp(exp_decay, (0.0, 4.0), y0, method=method,
                              rtol=1e-8, atol=1e-10)
    assert res.success
    expect = y0 * np.exp(-0.5 * 4.0)
    assert np.allclose(res.y[:, -1], expect, rtol=1e-6)


@pytest.mark.parametrize("method", ["RK45", "DOP853"])
def test_matches_scipy_trajectory(method):
    rng = np.random.default_rng(0)
    n = 12
    M = rng.random((n, n)) - 0.5
    M = M - np.eye(n) * 2

    def f_ours(t, y):
        import sparse

        return sparse.asdistarray(M @ np.asarray(y))

    def f_sp(t, y):
        return M @ y

    y0 = rng.random(n)
    ours = integrate.solve_ivp(f_ours, (0, 2.0), y0, method=method,
                               rtol=1e-9, atol=1e-11)
    ref = sp_solve_ivp(f_sp, (0, 2.0), y0, method=method, rtol=1e-9, atol=1e-11)
    assert ours.success and ref.success
    assert np.allclose(ours.y[:, -1], ref.y[:, -1], rtol=1e-6, atol=1e-9)


def test_dop853_step_count_matches_scipy():
    """The 8(5,3) error norm must use the un-normalized squared sums
    (|h|*s5/sqrt((s5+0.01*s3)*n)); a 1/sqrt(n)-loose norm accepts far
    fewer/larger steps than scipy at the same rtol."""
    n = 400
    y0 = np.linspace(1.0, 2.0, n)
    ours = integrate.solve_ivp(exp_decay, (0.0, 10.0), y0, method="DOP853",
                               rtol=1e-10, atol=1e-12)
    ref = sp_solve_ivp(lambda t, y: -0.5 * y, (0.0, 10.0), y0,
                       method="DOP853", rtol=1e-10, atol=1e-12)
    assert ours.success and ref.success
    # same step controller => step counts within 25% of each other
    assert abs(len(ours.t) - len(ref.t)) <= max(3, 0.25 * len(ref.t))
    assert np.allclose(ours.y[:, -1], ref.y[:, -1], rtol=1e-8)


def test_t_eval_and_dense_output():
    y0 = np.array([1.0])
    t_eval = np.linspace(0, 3, 7)
    res = integrate.solve_ivp(exp_decay, (0, 3.0), y0, t_eval=t_eval,
                              dense_output=True, rtol=1e-9, atol=1e-12)
    assert res.y.shape == (1, 7)
    assert np.allclose(res.y[0], np.exp(-0.5 * t_eval), rtol=1e-6)
    mid = res.sol(1.234)
    assert np.isclose(np.asarray(mid)[0], np.exp(-0.5 * 1.234), rtol=1e-6)


def test_events():
    def hit(t, y):
        return float(np.asarray(y)[0]) - 0.5

    hit.terminal = True
    hit.direction = -1
    res = integrate.solve_ivp(exp_decay, (0, 20.0), np.array([1.0]),
                              events=hit, rtol=1e-9, atol=1e-12)
    assert res.status == 1
    te = res.t_events[0][0]
    assert np.isclose(te, np.log(2.0) / 0.5, rtol=1e-5)


def test_sparse_rhs():
    """Hamiltonian-style RHS: dy/dt = -A y with A sparse (the quantum-app
    shape, reference integrate usage)."""
    import scipy.sparse as sps

    import sparse
    from sparse import csr_array

    n = 30
    A = sps.random(n, n, 0.2, random_state=1)
    A = (A + A.T).tocsr() * 0.1

    def f(t, y):
        return csr_array(A).dot(sparse.asdistarray(y)) * (-1.0)

    y0 = np.ones(n)
    res = integrate.solve_ivp(f, (0, 1.0), y0, method="RK45", rtol=1e-8,
                              atol=1e-10)
    ref = sp_solve_ivp(lambda t, y: -(A @ y), (0, 1.0), y0, method="RK45",
                       rtol=1e-8, atol=1e-10)
    assert np.allclose(res.y[:, -1], ref.y[:, -1], rtol=1e-5)
