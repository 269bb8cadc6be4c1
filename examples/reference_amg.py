"""Host-side (scipy-only) smoothed-aggregation AMG reference.

Capability parity with reference examples/reference_amg.py, which builds a
pyamg smoothed_aggregation_solver as the CPU oracle for examples/amg.py.
pyamg is optional here: when installed we use it (same knobs as the
reference — Jacobi omega=4/3, one sweep, no candidate improvement);
otherwise a self-contained scipy SA-AMG (symmetric strength, greedy
aggregation, Jacobi-smoothed tentative prolongator, Galerkin coarsening)
provides the same role.  Usage:

    python examples/reference_amg.py -n 65536 [-theta 0.08] [-maxiter 200]
"""
import argparse
import math
import time

import numpy as np
import scipy.sparse as sps
import scipy.sparse.linalg as spla


def strength(A: sps.csr_matrix, theta: float) -> sps.csr_matrix:
    """Symmetric strength-of-connection: keep |a_ij| >= theta*sqrt(|a_ii a_jj|)."""
    d = np.abs(A.diagonal())
    C = A.tocoo(copy=True)
    keep = np.abs(C.data) >= theta * np.sqrt(d[C.row] * d[C.col])
    keep |= C.row == C.col
    return sps.csr_matrix((C.data[keep], (C.row[keep], C.col[keep])), A.shape)


def aggregate(C: sps.csr_matrix) -> np.ndarray:
    """Greedy distance-1 aggregation (pyamg 'standard' style): pass 1 seeds
    aggregates from fully-unaggregated neighborhoods, pass 2 attaches the
    rest to a neighboring aggregate."""
    n = C.shape[0]
    agg = -np.ones(n, dtype=np.int64)
    next_agg = 0
    indptr, indices = C.indptr, C.indices
    for i in range(n):
        if agg[i] != -1:
            continue
        nbrs = indices[indptr[i]: indptr[i + 1]]
        if np.all(agg[nbrs] == -1):
            agg[i] = next_agg
            agg[nbrs] = next_agg
            next_agg += 1
    for i in range(n):
        if agg[i] == -1:
            nbrs = indices[indptr[i]: indptr[i + 1]]
            owned = nbrs[agg[nbrs] != -1]
            agg[i] = agg[owned[0]] if owned.size else next_agg
            if not owned.size:
                next_agg += 1
    return agg


def build_hierarchy(A: sps.csr_matrix, theta: float = 0.0,
                    max_coarse: int = 512, omega: float = 4.0 / 3.0):
    levels = []
    while A.shape[0] > max_coarse and len(levels) < 20:
        C = strength(A, theta) if theta > 0 else A
        agg = aggregate(C)
        nc = int(agg.max()) + 1
        if nc >= A.shape[0]:
            break
        T = sps.csr_matrix(
            (np.ones(A.shape[0]), (np.arange(A.shape[0]), agg)),
            shape=(A.shape[0], nc))
        # Jacobi-smoothed prolongator: P = (I - omega D^-1 A) T
        dinv = 1.0 / A.diagonal()
        P = (T - sps.diags(omega * dinv) @ (A @ T)).tocsr()
        R = P.T.tocsr()
        levels.append(dict(A=A, P=P, R=R, dinv=dinv))
        A = (R @ A @ P).tocsr()
    levels.append(dict(A=A, lu=spla.splu(A.tocsc())))
    return levels


def vcycle(levels, li, b):
    lvl = levels[li]
    if "lu" in lvl:
        return lvl["lu"].solve(b)
    A, dinv, w = lvl["A"], lvl["dinv"], 2.0 / 3.0
    x = w * dinv * b
    r = b - A @ x
    x += lvl["P"] @ vcycle(levels, li + 1, lvl["R"] @ r)
    r = b - A @ x
    x += w * dinv * r
    return x


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("-n", type=int, default=65536)
    ap.add_argument("-theta", type=float, default=0.0)
    ap.add_argument("-maxiter", type=int, default=200)
    ap.add_argument("-tol", type=float, default=1e-8)
    args = ap.parse_args()

    nx = int(round(math.sqrt(args.n)))
    n = nx * nx
    A = (sps.kron(sps.eye(nx), sps.diags([-1, 2, -1], [-1, 0, 1], (nx, nx)))
         + sps.kron(sps.diags([-1, 2, -1], [-1, 0, 1], (nx, nx)),
                    sps.eye(nx))).tocsr()
    b = np.ones(n)

    t0 = time.time()
    try:
        import pyamg

        smoother = ("jacobi", {"omega": 4.0 / 3.0, "iterations": 1})
        ml = pyamg.aggregation.smoothed_aggregation_solver(
            A, keep=True, improve_candidates=None, presmoother=smoother,
            postsmoother=smoother)
        M = ml.aspreconditioner()
        src = "pyamg"
    except ImportError:
        levels = build_hierarchy(A, theta=args.theta)
        M = spla.LinearOperator(A.shape, matvec=lambda r: vcycle(levels, 0, r))
        src = "scipy SA-AMG (pyamg not installed)"
    setup_s = time.time() - t0

    iters = [0]
    t0 = time.time()
    x, info = spla.cg(A, b, M=M, maxiter=args.maxiter,
                      rtol=args.tol, atol=0.0,
                      callback=lambda xk: iters.__setitem__(0, iters[0] + 1))
    solve_s = time.time() - t0
    r = np.linalg.norm(b - A @ x)
    print(f"[{src}] n={n} setup={setup_s * 1e3:.1f}ms "
          f"solve={solve_s * 1e3:.1f}ms iters={iters[0]} "
          f"({iters[0] / max(solve_s, 1e-9):.2f} iters/s) residual={r:.3e} "
          f"info={info}")


if __name__ == "__main__":
    main()
