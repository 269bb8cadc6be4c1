"""Shared fixtures: scipy-oracle matrices and .mtx files.

Mirrors the role of the reference's tests/integration/utils/common.py
(fixture list + dtype sweep) with our own generated fixtures.
"""
import os

import numpy as np
import scipy.sparse as sps

types = [np.float32, np.float64, np.complex64, np.complex128]
real_types = [np.float32, np.float64]

_TESTDATA = os.path.join(os.path.dirname(__file__), "..", "testdata")


def _ensure_testdata():
    os.makedirs(_TESTDATA, exist_ok=True)
    test = os.path.join(_TESTDATA, "test.mtx")
    if not os.path.exists(test):
        with open(test, "w") as f:
            f.write("%%MatrixMarket matrix coordinate real general\n")
            f.write("5 5 7\n")
            for (i, j, v) in [(1, 1, 2.0), (1, 4, -1.5), (2, 2, 3.25),
                              (3, 1, 0.5), (3, 3, 1.0), (4, 5, -2.0),
                              (5, 2, 4.0)]:
                f.write(f"{i} {j} {v}\n")
    sym = os.path.join(_TESTDATA, "sym.mtx")
    if not os.path.exists(sym):
        with open(sym, "w") as f:
            f.write("%%MatrixMarket matrix coordinate real symmetric\n")
            f.write("4 4 5\n")
            for (i, j, v) in [(1, 1, 4.0), (2, 1, -1.0), (3, 2, -1.0),
                              (4, 4, 4.0), (4, 3, 0.5)]:
                f.write(f"{i} {j} {v}\n")
    pat = os.path.join(_TESTDATA, "pattern.mtx")
    if not os.path.exists(pat):
        with open(pat, "w") as f:
            f.write("%%MatrixMarket matrix coordinate pattern general\n")
            f.write("3 4 4\n")
            for (i, j) in [(1, 1), (2, 3), (3, 2), (3, 4)]:
                f.write(f"{i} {j}\n")
    rnd = os.path.join(_TESTDATA, "rand33.mtx")
    if not os.path.exists(rnd):
        m = sps.random(33, 29, density=0.11, random_state=5)
        import scipy.io as sio

        sio.mmwrite(rnd, m)
        # scipy may write array or .mtx.gz; force plain path
    return _TESTDATA


def mtx_files():
    d = _ensure_testdata()
    return [os.path.join(d, f) for f in ("test.mtx", "sym.mtx", "rand33.mtx")]


test_mtx_files = mtx_files()
