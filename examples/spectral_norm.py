"""Spectral norm via power iteration (capability parity with reference
examples/spectral_norm.py)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

from sparse import csr_array, darray


def normest(M, tol=1e-4, max_it=10):
    """2-norm approximation by the power method (works for csr_array or
    numpy)."""
    is_ours = isinstance(M, csr_array)
    rng = np.random.default_rng(15210)
    x = rng.random(M.shape[1])
    if is_ours:
        xd = darray.asdistarray(x)
        y = M.dot(xd)
        pnorm = float(y.norm().item())
        x = y * (1.0 / pnorm)
        res, it = 1.0, 0
        while res > tol and it < max_it:
            y = M.dot(x)
            ynorm = float(y.norm().item())
            res = abs(pnorm - ynorm)
            pnorm = ynorm
            x = y * (1.0 / ynorm)
            it += 1
        v = M.dot(x)
        return float(v.norm().item())
    y = M @ x
    pnorm = np.linalg.norm(y)
    x = y / pnorm
    res, it = 1.0, 0
    while res > tol and it < max_it:
        y = M @ x
        ynorm = np.linalg.norm(y)
        res = abs(pnorm - ynorm)
        pnorm = ynorm
        x = y / ynorm
        it += 1
    return np.linalg.norm(M @ x)


if __name__ == "__main__":
    rng = np.random.default_rng(15210)
    M = rng.random((100, 100))
    A = csr_array(M)
    ours, ref = normest(A), normest(M)
    print(f"normest(csr)={ours:.6f} normest(dense)={ref:.6f}")
    assert np.isclose(ours, ref, rtol=1e-3)
    print("OK")
