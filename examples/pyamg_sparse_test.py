"""pyamg integration smoke test (capability parity with reference
examples/pyamg_legate_test.py): build a pyamg smoothed-aggregation
hierarchy on the host, run the preconditioned CG through this framework.
Skips cleanly when pyamg is not installed.

python examples/pyamg_sparse_test.py -n 65536
"""
import argparse
import math
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import numpy as np
import scipy.sparse as sps

try:
    import pyamg
except ImportError:
    print("pyamg not installed; skipping (install pyamg to run this test)")
    sys.exit(0)

from pyamg_to_sparse import patch  # noqa: E402

from sparse import csr_array, linalg  # noqa: E402

ap = argparse.ArgumentParser()
ap.add_argument("-n", type=int, default=65536)
ap.add_argument("-maxiter", type=int, default=200)
args = ap.parse_args()

nx = int(round(math.sqrt(args.n)))
n = nx * nx
A = (sps.kron(sps.eye(nx), sps.diags([-1, 2, -1], [-1, 0, 1], (nx, nx)))
     + sps.kron(sps.diags([-1, 2, -1], [-1, 0, 1], (nx, nx)),
                sps.eye(nx))).tocsr()
b = np.ones(n)

patch(pyamg)
smoother = ("jacobi", {"omega": 4.0 / 3.0, "iterations": 1})
ml = pyamg.aggregation.smoothed_aggregation_solver(
    A, keep=True, improve_candidates=None, presmoother=smoother,
    postsmoother=smoother)
M = ml.aspreconditioner()  # patched: MI355X-backed V-cycle

iters = [0]
t0 = time.time()
x, info = linalg.cg(csr_array(A), b, M=M, maxiter=args.maxiter, tol=1e-8,
                    callback=lambda xk: iters.__setitem__(0, iters[0] + 1),
                    conv_test_iters=2)
solve_s = time.time() - t0
r = np.linalg.norm(b - A @ np.asarray(x))
print(f"n={n} iters={iters[0]} ({iters[0] / max(solve_s, 1e-9):.1f} iters/s) "
      f"residual={r:.3e} info={info}")
