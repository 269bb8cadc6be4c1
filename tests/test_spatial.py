"""cdist (coverage parity: reference test_spatial.py)."""
import numpy as np

from sparse import spatial

from utils.sample import sample_dense


def test_cdist():
    XA = sample_dense((20, 5), seed=1)
    XB = sample_dense((13, 5), seed=2)
    D = spatial.cdist(XA, XB)
    from scipy.spatial.distance import cdist as sp_cdist

    assert np.allclose(np.asarray(D), sp_cdist(XA, XB), atol=1e-10)


def test_cdist_same_points():
    X = sample_dense((9, 3), seed=3)
    D = np.asarray(spatial.cdist(X, X))
    assert np.allclose(np.diag(D), 0.0, atol=1e-7)
