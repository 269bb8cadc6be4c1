"""Run pyamg-built hierarchies on the MI355X sparse framework.

Capability parity with reference examples/pyamg_to_legate/wrapper.py:
`patch(pyamg)` redirects a pyamg MultilevelSolver's preconditioner
application to this framework (device csr_arrays, fused Jacobi smoother,
hipGraph-captured V-cycle), so existing pyamg scripts accelerate without
modification; `from_pyamg(ml)` does the conversion explicitly.
"""
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from sparse import csr_array, darray, linalg  # noqa: E402


def _level_from(Alvl):
    A = csr_array(Alvl.tocsr())
    d = A.diagonal()
    return A, darray.DistArray.from_local(1.0 / d.local, d.partition, d.shape)


def from_pyamg(ml):
    """Convert a pyamg MultilevelSolver into this framework's level list
    (same shape as examples/amg.py build_hierarchy output) and return a
    LinearOperator applying one V(1,1) cycle with weighted-Jacobi
    smoothing on the GPU."""
    import torch

    from amg import vcycle  # examples/amg.py

    levels = []
    for i, lvl in enumerate(ml.levels[:-1]):
        A, dinv = _level_from(lvl.A)
        levels.append({
            "A": A,
            "P": csr_array(lvl.P.tocsr()),
            "R": csr_array(lvl.R.tocsr()),
            "dinv": dinv,
            "omega": 2.0 / 3.0,
        })
    Ac = csr_array(ml.levels[-1].A.tocsr())
    coarse = ml.levels[-1].A.toarray()
    levels.append({"A": Ac, "coarse_inv": torch.as_tensor(
        np.linalg.pinv(coarse), device=Ac._values.device,
        dtype=Ac._values.dtype)})

    n = levels[0]["A"].shape[0]

    def M(r, out=None):
        x = vcycle(levels, 0, darray.asdistarray(r))
        if out is not None:
            out.local.copy_(x.local)
            return out
        return x

    op = linalg.LinearOperator((n, n), matvec=M, dtype=np.float64)
    op.levels = levels
    return op


def patch(pyamg):
    """Monkeypatch pyamg so MultilevelSolver.aspreconditioner() returns the
    MI355X-backed V-cycle operator (reference wrapper.patch parity)."""
    Ml = pyamg.multilevel.MultilevelSolver

    def aspreconditioner(self, cycle="V"):
        return from_pyamg(self)

    Ml.aspreconditioner = aspreconditioner
    return pyamg
