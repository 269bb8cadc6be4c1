"""CSR add/sub/multiply (coverage parity: reference test_csr_elemwise.py)."""
import numpy as np
import pytest

from sparse import csr_array

from utils.common import types
from utils.sample import sample_csr, sample_dense


@pytest.mark.parametrize("dt", types)
def test_add(dt):
    a = sample_csr(15, 11, 0.3, seed=1, dtype=dt)
    b = sample_csr(15, 11, 0.3, seed=2, dtype=dt)
    c = csr_array(a) + csr_array(b)
    assert np.allclose(np.asarray(c.todense()), (a + b).toarray())


@pytest.mark.parametrize("dt", types)
def test_sub(dt):
    a = sample_csr(15, 11, 0.3, seed=3, dtype=dt)
    b = sample_csr(15, 11, 0.3, seed=4, dtype=dt)
    c = csr_array(a) - csr_array(b)
    assert np.allclose(np.asarray(c.todense()), (a - b).toarray())


@pytest.mark.parametrize("dt", types)
def test_elem_mult(dt):
    a = sample_csr(13, 17, 0.4, seed=5, dtype=dt)
    b = sample_csr(13, 17, 0.4, seed=6, dtype=dt)
    c = csr_array(a).multiply(csr_array(b))
    assert np.allclose(np.asarray(c.todense()), (a.multiply(b)).toarray())


@pytest.mark.parametrize("dt", types)
def test_mult_dense(dt):
    a = sample_csr(9, 12, 0.5, seed=7, dtype=dt)
    d = sample_dense((9, 12), seed=8, dtype=dt)
    c = csr_array(a).multiply(d)
    assert np.allclose(np.asarray(c.todense()), a.multiply(d).toarray())


def test_mult_scalar_and_div():
    a = sample_csr(8, 8, 0.5, seed=9)
    A = csr_array(a)
    assert np.allclose(np.asarray((A * 2.5).todense()), (a * 2.5).toarray())
    assert np.allclose(np.asarray((2.5 * A).todense()), (a * 2.5).toarray())
    assert np.allclose(np.asarray((A / 2.0).todense()), (a / 2.0).toarray())
    assert np.allclose(np.asarray((-A).todense()), (-a).toarray())


def test_add_mismatched_structure():
    a = sample_csr(10, 10, 0.2, seed=10)
    b = sample_csr(10, 10, 0.6, seed=11)
    assert np.allclose(np.asarray((csr_array(a) + csr_array(b)).todense()),
                       (a + b).toarray())


def test_dtype_promotion():
    a = sample_csr(7, 7, 0.4, seed=12, dtype=np.float32)
    b = sample_csr(7, 7, 0.4, seed=13, dtype=np.float64)
    c = csr_array(a) + csr_array(b)
    assert c.dtype == np.float64
