"""Iterative solvers on DistArrays.

Reference parity: sparse/linalg.py — LinearOperator protocol with out=
(linalg.py:128-355), fused cg_axpby keeping scalars as device tensors
(linalg.py:479-496 + axpby.cu:25-42), cg with preconditioner/callback and
deferred convergence checks every conv_test_iters iterations
(linalg.py:499-565), cgs (:570-615), bicg (:620-665), gmres restarted
Arnoldi with host-side lstsq (:670-792), bicgstab (:795-... — fixed here;
the reference's is marked broken), lsqr Golub-Kahan (:937-1415), eigsh
thick-restart Lanczos (:1416-1569), spsolve=cg (:88-122).

Asynchrony discipline (SURVEY §7 hard parts): every scalar produced by a
reduction stays a 0-dim device tensor; the host blocks only at the batched
convergence checks.
"""
from __future__ import annotations

import math
import os
from typing import Callable, Optional

import numpy as np
import torch

from .coverage import track_provenance
from .darray import DistArray, asdistarray
from . import darray
from .module import is_sparse_matrix
from .parallel import comm

__all__ = [
    "LinearOperator", "IdentityOperator", "aslinearoperator", "cg", "cgs",
    "bicg", "bicgstab", "gmres", "minres", "lsqr", "eigsh", "svds",
    "spsolve", "cg_axpby", "norm",
]


def norm(A, ord="fro", axis=None):
    """Matrix norms of a sparse array (scipy.sparse.linalg.norm parity —
    a superset of the reference, which exposes no norm).  Supports
    'fro' (default), 1, inf, and elementwise axis=None variants."""
    if axis is not None:
        raise NotImplementedError("axis norms are not implemented")
    v = A._values_tensor()
    if ord in ("fro", None):
        local = (v.abs() ** 2).sum() if v.numel() else torch.zeros(
            (), dtype=torch.float64)
        local = local.real.double() if local.is_complex() else local.double()
        comm.all_reduce_(local)
        return float(local.item()) ** 0.5
    if ord == 1:  # max column sum of |A|
        col = abs(A).sum(axis=0)
        return float(np.max(np.asarray(col)))
    if ord in (np.inf, float("inf"), "inf"):
        row = abs(A).sum(axis=1)
        return float(np.max(np.asarray(row)))
    raise ValueError(f"norm ord {ord!r} not supported")


# -- fused axpby --------------------------------------------------------------
def cg_axpby(y: DistArray, x: DistArray, a, b, isalpha: bool = True,
             negate: bool = False) -> DistArray:
    """y = y ± (a/b)·x  (isalpha) or  y = x ± (a/b)·y, a/b 0-dim tensors.

    Reference: the AXPBY task (linalg.py:479-496, axpby.cu:25-42) — fused so
    the scalar quotient never touches the host."""
    at = a if isinstance(a, torch.Tensor) else torch.as_tensor(a, device=y.local.device)
    bt = b if isinstance(b, torch.Tensor) else torch.as_tensor(b, device=y.local.device)
    if y.local.is_cuda:
        from . import kernels

        kernels.require()
        kernels.axpby(y.local, x.local, at.to(y.local.dtype), bt.to(y.local.dtype),
                      isalpha, negate)
        return y
    s = at / bt
    if negate:
        s = -s
    if isalpha:
        y.local.add_(x.local * s)
    else:
        y.local.mul_(s).add_(x.local)
    return y


def _axpby_norm2(y: DistArray, x: DistArray, a, b, negate: bool) -> torch.Tensor:
    """Fused y += ±(a/b)x and all-reduced sum(y_new^2) (GPU real dtypes)."""
    from . import kernels

    dot = kernels.axpby_norm2(y.local, x.local, a.to(y.local.dtype),
                              b.to(y.local.dtype), True, negate)
    comm.all_reduce_(dot)
    return dot


# -- operators ----------------------------------------------------------------
class LinearOperator:
    def __init__(self, shape, matvec: Optional[Callable] = None, rmatvec=None,
                 dtype=None):
        self.shape = tuple(shape)
        self.dtype = np.dtype(dtype) if dtype is not None else np.dtype(np.float64)
        self._matvec_fn = matvec
        self._rmatvec_fn = rmatvec

    def matvec(self, x, out=None):
        if self._matvec_fn is None:
            raise NotImplementedError
        try:
            return self._matvec_fn(x, out=out)
        except TypeError:
            r = self._matvec_fn(x)
            r = asdistarray(r)
            if out is not None:
                out.local.copy_(r.local.to(out.local.dtype))
                return out
            return r

    def rmatvec(self, x, out=None):
        if self._rmatvec_fn is None:
            raise NotImplementedError
        try:
            return self._rmatvec_fn(x, out=out)
        except TypeError:
            r = asdistarray(self._rmatvec_fn(x))
            if out is not None:
                out.local.copy_(r.local.to(out.local.dtype))
                return out
            return r

    def __matmul__(self, x):
        return self.matvec(x)


class IdentityOperator(LinearOperator):
    """Reference linalg.py:437-459."""

    def __init__(self, shape, dtype=None):
        super().__init__(shape, dtype=dtype)

    def matvec(self, x, out=None):
        x = asdistarray(x)
        if out is not None:
            out.local.copy_(x.local.to(out.local.dtype))
            return out
        return x.copy()

    rmatvec = matvec


class _SparseMatrixLinearOperator(LinearOperator):
    """Reference linalg.py:420-432."""

    def __init__(self, A):
        self.A = A
        self._AH = None  # adjoint, built lazily (conjugate for complex)
        super().__init__(A.shape, dtype=A.dtype)

    def matvec(self, x, out=None):
        return self.A.dot(asdistarray(x), out=out)

    def rmatvec(self, x, out=None):
        if self._AH is None:
            self._AH = (self.A.conj().T if np.issubdtype(self.A.dtype,
                        np.complexfloating) else self.A.T)
        return self._AH.dot(asdistarray(x), out=out)


def aslinearoperator(A) -> LinearOperator:
    if isinstance(A, LinearOperator):
        return A
    if is_sparse_matrix(A):
        return _SparseMatrixLinearOperator(A)
    if isinstance(A, (np.ndarray, torch.Tensor, DistArray)):
        Ad = asdistarray(A)

        def mv(x, out=None):
            xg = asdistarray(x).gather()
            r = Ad.local @ xg
            res = DistArray.from_local(r, Ad.partition, (Ad.shape[0],))
            if out is not None:
                out.local.copy_(res.local.to(out.local.dtype))
                return out
            return res

        def rmv(x, out=None):
            xg = asdistarray(x).gather()
            me = comm.rank()
            part = Ad.partition
            r = Ad.local.conj().T @ xg[part.start(me): part.stop(me)]
            comm.all_reduce_(r)
            res = DistArray.from_global(r)
            if out is not None:
                out.local.copy_(res.local.to(out.local.dtype))
                return out
            return res

        return LinearOperator(Ad.shape, matvec=mv, rmatvec=rmv, dtype=Ad.dtype)
    raise TypeError(f"cannot make a LinearOperator from {type(A)}")


make_linear_operator = aslinearoperator


def _vec(b, dtype=None) -> DistArray:
    v = asdistarray(b)
    if dtype is not None and v.dtype != dtype:
        v = v.astype(dtype)
    return v


def _tols(bnorm: float, tol: float, atol) -> float:
    if atol is None:
        atol = 0.0
    return max(float(tol) * bnorm, float(atol))


# -- CG -----------------------------------------------------------------------
@track_provenance(nested=True)
def cg(A, b, x0=None, tol=1e-5, maxiter=None, M=None, callback=None, atol=None,
       conv_test_iters=25):
    """Preconditioned conjugate gradients (reference linalg.py:499-565)."""
    A = aslinearoperator(A)
    b = _vec(b, A.dtype)
    n = b.shape[0]
    if maxiter is None:
        maxiter = n * 10
    ident_M = M is None or isinstance(M, IdentityOperator)
    M = aslinearoperator(M) if M is not None else IdentityOperator(A.shape, dtype=A.dtype)
    x = _vec(x0, A.dtype).copy() if x0 is not None else darray.zeros((n,), dtype=A.dtype)
    r = b - A.matvec(x)
    # with an identity preconditioner z IS r — no per-iteration copy
    z = r if ident_M else M.matvec(r)
    p = z.copy()
    rz = r.dot(z)
    q = darray.zeros((n,), dtype=A.dtype)
    bnorm = float(b.norm().item())
    if bnorm == 0.0:
        bnorm = 1.0
    threshold = _tols(bnorm, tol, atol)
    info = maxiter
    # fully-fused GPU path: SpMV+p·Ap in one kernel, r-update+|r|^2 in one
    # kernel — the MI355X realization of the reference's future-based
    # asynchrony (3 HBM passes per iteration beyond the SpMV).
    fused = (ident_M and isinstance(A, _SparseMatrixLinearOperator)
             and getattr(A.A, "_format", None) == "csr"
             and b.local.is_cuda and not b.local.is_complex())
    use_cg2 = fused and os.environ.get("SPARSE_CG2") == "1"
    if use_cg2 and comm.world_size() > 1:
        # the DIA mirror can build on some ranks and not others (memory
        # headroom, non-uniform slabs); a rank-divergent branch choice means
        # divergent collective sequences → deadlock.  All-reduce(min) the flag.
        flag = torch.tensor(
            [1.0 if A.A._dia() is not None else 0.0], device=b.local.device)
        comm.all_reduce_(flag, op="min")
        use_cg2 = bool(flag.item() > 0.5)
    elif use_cg2:
        use_cg2 = A.A._dia() is not None
    if use_cg2:
        # two-kernel CG iteration on the DIA fast path: K1 folds the
        # p-update into the SpMV (p = r + beta p, q = Ap, p.q), K2 fuses
        # x += alpha p, r -= alpha q and |r|^2.  MEASURED SLOWER than the
        # 4-kernel loop on MI355X at nx=16384 (6.14 vs 5.89 ms/iter:
        # K1 streams TWO vector windows through L2 and is issue-bound at
        # 5.0 TB/s while the axpby passes it replaces run at 6.0), so this
        # is opt-in via SPARSE_CG2=1; kept because the tradeoff may flip
        # for wider stencils (more dvals reuse per vector byte).
        from . import kernels

        p_b = darray.zeros((n,), dtype=A.dtype)
        bufs = (p, p_b)
        rz_cur, rz_old = rz, rz
        for i in range(maxiter):
            if i == 0:
                pq = A.A.spmv_dot(p, q)
                cur = p
            else:
                # p_old alternates: iter 1 reads p writes p_b, iter 2 reads
                # p_b writes p, ...
                pold, pnew = bufs[(i + 1) % 2], bufs[i % 2]
                pq = A.A.spmv_bpdot(r, pold, pnew, q, rz_cur, rz_old)
                cur = pnew
            rz_new = kernels.cg_xr_norm2(x.local, cur.local, r.local,
                                         q.local, rz_cur, pq)
            comm.all_reduce_(rz_new)
            rz_old, rz_cur = rz_cur, rz_new
            if conv_test_iters and (i % conv_test_iters == 0 or i == maxiter - 1):
                if math.sqrt(max(float(rz_cur.item()), 0.0)) < threshold:
                    info = 0
                    break
            if callback is not None:
                callback(x)
        if info != 0 and float(r.norm().item()) < threshold:
            info = 0
        return x, info
    for i in range(maxiter):
        if fused:
            from . import kernels

            # 3-kernel iteration: SpMV+p·Ap fused; x/r updates + |r|^2
            # fused (cg_xr_norm2); p-update axpby.  Same HBM bytes as the
            # 4-kernel form (measured parity at nx=16384), one launch and
            # one Python dispatch fewer per iteration — which is what
            # matters in the eager ws>1 loop.  SPARSE_CG4=1 in bench.py
            # keeps the 4-kernel formulation for A/B.
            pq = A.A.spmv_dot(p, q)
            rz_new = kernels.cg_xr_norm2(x.local, p.local, r.local, q.local,
                                         rz, pq)
            comm.all_reduce_(rz_new)
            if conv_test_iters and (i % conv_test_iters == 0 or i == maxiter - 1):
                if math.sqrt(max(float(rz_new.item()), 0.0)) < threshold:
                    info = 0
                    break
            cg_axpby(p, r, rz_new, rz, isalpha=False, negate=False)
            rz = rz_new
            if callback is not None:
                callback(x)
            continue
        A.matvec(p, out=q)
        pq = p.dot(q)
        # x += (rz/pq) p ; r -= (rz/pq) q — fused, scalars stay on device
        cg_axpby(x, p, rz, pq, isalpha=True, negate=False)
        cg_axpby(r, q, rz, pq, isalpha=True, negate=True)
        if conv_test_iters and (i % conv_test_iters == 0 or i == maxiter - 1):
            if float(r.norm().item()) < threshold:
                info = 0
                break
        if not ident_M:
            M.matvec(r, out=z)
        rz_new = r.dot(z)
        # p = z + (rz_new/rz) p
        cg_axpby(p, z, rz_new, rz, isalpha=False, negate=False)
        rz = rz_new
        if callback is not None:
            callback(x)
    else:
        info = maxiter
    if info != 0 and float(r.norm().item()) < threshold:
        info = 0
    return x, info


def spsolve(A, b, **kwargs):
    """Plain CG solve (reference linalg.py:88-122)."""
    x, _ = cg(A, b, tol=1e-10, **kwargs)
    return x


# -- CGS ----------------------------------------------------------------------
@track_provenance(nested=True)
def cgs(A, b, x0=None, tol=1e-5, maxiter=None, M=None, callback=None, atol=None,
        conv_test_iters=25):
    """Preconditioned conjugate gradient squared (reference
    linalg.py:570-615; M semantics per scipy.sparse.linalg.cgs)."""
    A = aslinearoperator(A)
    b = _vec(b, A.dtype)
    n = b.shape[0]
    maxiter = maxiter or n * 10
    ident_M = M is None or isinstance(M, IdentityOperator)
    Mop = aslinearoperator(M) if not ident_M else None
    x = _vec(x0, A.dtype).copy() if x0 is not None else darray.zeros((n,), dtype=A.dtype)
    r = b - A.matvec(x)
    rtilde = r.copy()
    bnorm = float(b.norm().item()) or 1.0
    threshold = _tols(bnorm, tol, atol)
    rho = None
    info = maxiter
    for i in range(maxiter):
        rho_new = rtilde.dot(r)
        if i == 0:
            u = r.copy()
            p = r.copy()
        else:
            beta = rho_new / rho
            u = r + q_ * beta
            p = u + (q_ + p * beta) * beta
        rho = rho_new
        phat = p if ident_M else Mop.matvec(p)
        vhat = A.matvec(phat)
        sigma = rtilde.dot(vhat)
        alpha = rho / sigma
        q_ = u - alpha * vhat
        uq = u + q_
        uqhat = uq if ident_M else Mop.matvec(uq)
        x += uqhat * alpha
        r -= A.matvec(uqhat) * alpha
        if conv_test_iters and (i % conv_test_iters == 0 or i == maxiter - 1):
            if float(r.norm().item()) < threshold:
                info = 0
                break
        if callback is not None:
            callback(x)
    if info != 0 and float(r.norm().item()) < threshold:
        info = 0
    return x, info


# -- BiCG ---------------------------------------------------------------------
@track_provenance(nested=True)
def bicg(A, b, x0=None, tol=1e-5, maxiter=None, M=None, callback=None, atol=None,
         conv_test_iters=25):
    """Preconditioned biconjugate gradients (reference linalg.py:620-665;
    M semantics per scipy.sparse.linalg.bicg — z = M r, ztilde = M^H
    rtilde)."""
    A = aslinearoperator(A)
    b = _vec(b, A.dtype)
    n = b.shape[0]
    maxiter = maxiter or n * 10
    ident_M = M is None or isinstance(M, IdentityOperator)
    Mop = aslinearoperator(M) if not ident_M else None
    x = _vec(x0, A.dtype).copy() if x0 is not None else darray.zeros((n,), dtype=A.dtype)
    r = b - A.matvec(x)
    rtilde = r.copy()
    z = r if ident_M else Mop.matvec(r)
    ztilde = rtilde if ident_M else Mop.rmatvec(rtilde)
    p = z.copy()
    ptilde = ztilde.copy()
    rho = rtilde.dot(z)
    bnorm = float(b.norm().item()) or 1.0
    threshold = _tols(bnorm, tol, atol)
    info = maxiter
    for i in range(maxiter):
        q = A.matvec(p)
        qtilde = A.rmatvec(ptilde)
        alpha = rho / ptilde.dot(q)
        x += p * alpha
        r -= q * alpha
        rtilde -= qtilde * alpha.conj()
        if conv_test_iters and (i % conv_test_iters == 0 or i == maxiter - 1):
            if float(r.norm().item()) < threshold:
                info = 0
                break
        z = r if ident_M else Mop.matvec(r)
        ztilde = rtilde if ident_M else Mop.rmatvec(rtilde)
        rho_new = rtilde.dot(z)
        beta = rho_new / rho
        rho = rho_new
        p = z + p * beta
        ptilde = ptilde * beta.conj() + ztilde
        if callback is not None:
            callback(x)
    if info != 0 and float(r.norm().item()) < threshold:
        info = 0
    return x, info


# -- BiCGSTAB -----------------------------------------------------------------
@track_provenance(nested=True)
def bicgstab(A, b, x0=None, tol=1e-5, maxiter=None, M=None, callback=None,
             atol=None, conv_test_iters=25):
    """BiCGSTAB (the reference's is commented 'Doesnt work',
    linalg.py:795-...; this follows the standard Van der Vorst form with a
    right preconditioner M per scipy: x-updates use the M-hatted vectors)."""
    A = aslinearoperator(A)
    b = _vec(b, A.dtype)
    n = b.shape[0]
    maxiter = maxiter or n * 10
    ident_M = M is None or isinstance(M, IdentityOperator)
    Mop = aslinearoperator(M) if not ident_M else None
    x = _vec(x0, A.dtype).copy() if x0 is not None else darray.zeros((n,), dtype=A.dtype)
    r = b - A.matvec(x)
    rhat = r.copy()
    bnorm = float(b.norm().item()) or 1.0
    threshold = _tols(bnorm, tol, atol)
    rho = alpha = omega = None
    v = p = None
    info = maxiter
    for i in range(maxiter):
        rho_new = rhat.dot(r)
        if i == 0:
            p = r.copy()
        else:
            beta = (rho_new / rho) * (alpha / omega)
            p = r + (p - v * omega) * beta
        rho = rho_new
        phat = p if ident_M else Mop.matvec(p)
        v = A.matvec(phat)
        alpha = rho / rhat.dot(v)
        s = r - v * alpha
        shat = s if ident_M else Mop.matvec(s)
        t = A.matvec(shat)
        omega = t.dot(s) / t.dot(t)
        x += phat * alpha + shat * omega
        r = s - t * omega
        if conv_test_iters and (i % conv_test_iters == 0 or i == maxiter - 1):
            if float(r.norm().item()) < threshold:
                info = 0
                break
        if callback is not None:
            callback(x)
    if info != 0 and float(r.norm().item()) < threshold:
        info = 0
    return x, info


class _Basis:
    """Device-resident Krylov basis (rows = basis vectors, local columns).

    Batched projections: ONE collective + ONE host sync per
    orthogonalization PASS instead of one device round-trip per basis
    vector (VERDICT r1 #10 — the asynchrony discipline of cg applied to
    the Arnoldi/Lanczos long tail).  All methods are SPMD-collective-safe:
    each rank calls with its local slab."""

    def __init__(self, nmax: int, template: DistArray):
        loc = template.local
        self.V = torch.zeros((nmax, loc.numel()), dtype=loc.dtype,
                             device=loc.device)
        self.n = 0
        self.part = template.partition
        self.vshape = template.shape

    def append(self, w_local: torch.Tensor) -> None:
        self.V[self.n].copy_(w_local)
        self.n += 1

    def row(self, j: int) -> DistArray:
        return DistArray.from_local(self.V[j], self.part, self.vshape)

    def project(self, w_local: torch.Tensor, j: int) -> torch.Tensor:
        """h = V[:j]^H w (globally reduced), one all-reduce."""
        Vj = self.V[:j]
        h = (Vj.conj() if Vj.is_complex() else Vj) @ w_local
        comm.all_reduce_(h)
        return h

    def deflate(self, w_local: torch.Tensor, h: torch.Tensor, j: int) -> None:
        """w -= V[:j]^T h (local update, no communication)."""
        w_local -= self.V[:j].transpose(0, 1) @ h

    def combine(self, coeff: torch.Tensor, j: int) -> torch.Tensor:
        """Σ_k coeff[k] V[k] as a local slab (no communication)."""
        return self.V[:j].transpose(0, 1) @ coeff.to(self.V.dtype)

    def global_norm(self, w_local: torch.Tensor) -> float:
        s = (w_local.conj() @ w_local).real if w_local.is_complex() \
            else w_local @ w_local
        comm.all_reduce_(s)
        return math.sqrt(max(float(s.item()), 0.0))


def svds(A, k=6, tol=0, maxiter=None, return_singular_vectors=True):
    """Largest-k singular triplets via thick-restart Lanczos on the Gram
    operator A^H A (scipy.sparse.linalg.svds surface; API superset — the
    reference has no SVD).  Returns (U, s, Vh) ascending in s like scipy."""
    Aop = aslinearoperator(A)
    m, n = Aop.shape
    if k <= 0 or k >= min(m, n):
        raise ValueError("k must be in (0, min(shape))")

    def gram(x, out=None):
        return Aop.rmatvec(Aop.matvec(x), out=out)

    G = LinearOperator((n, n), matvec=gram, dtype=Aop.dtype)
    w, V = eigsh(G, k=k, which="LA", tol=tol, maxiter=maxiter)
    s = np.sqrt(np.maximum(w, 0.0))
    order = np.argsort(s)  # scipy returns ascending
    s = s[order]
    V = V[:, order]
    if not return_singular_vectors:
        return s
    U = np.empty((m, k), dtype=V.dtype)
    for i in range(k):
        av = np.asarray(Aop.matvec(V[:, i]))
        U[:, i] = av / s[i] if s[i] > 0 else 0.0
    return U, s, V.conj().T


# -- MINRES -------------------------------------------------------------------
@track_provenance(nested=True)
def minres(A, b, x0=None, tol=1e-5, maxiter=None, M=None, callback=None,
           atol=None):
    """Minimum-residual method for symmetric (possibly indefinite) systems
    (Paige & Saunders Lanczos formulation; scipy.sparse.linalg.minres
    surface).  API superset of the reference, which has no indefinite
    solver — its CG diverges on saddle-point operators.

    The Lanczos recurrence consumes alfa/beta on the host each step (the
    Givens chain is inherently sequential), so this solver syncs ~3x per
    iteration — same as scipy; the asynchrony discipline of cg applies
    only to its fused path."""
    A = aslinearoperator(A)
    b = _vec(b, A.dtype)
    n = b.shape[0]
    maxiter = maxiter or 5 * n
    ident_M = M is None or isinstance(M, IdentityOperator)
    Mop = aslinearoperator(M) if not ident_M else None
    x = _vec(x0, A.dtype).copy() if x0 is not None else darray.zeros(
        (n,), dtype=A.dtype)
    r1 = b - A.matvec(x)
    y = r1 if ident_M else Mop.matvec(r1)
    beta1 = float(r1.dot(y).item())
    if beta1 < 0:
        raise ValueError("minres: preconditioner M is not positive definite")
    if beta1 == 0:
        return x, 0
    beta1 = math.sqrt(beta1)
    bnorm = float(b.norm().item()) or 1.0
    threshold = _tols(bnorm, tol, atol)
    oldb = 0.0
    beta = beta1
    dbar = epsln = phi = 0.0
    phibar = beta1
    cs = -1.0
    sn = 0.0
    w = darray.zeros((n,), dtype=A.dtype)
    w2 = darray.zeros((n,), dtype=A.dtype)
    r2 = r1
    info = maxiter
    for itn in range(1, maxiter + 1):
        s = 1.0 / beta
        v = y * s
        y = A.matvec(v)
        if itn >= 2:
            y = y - r1 * (beta / oldb)
        alfa = float(v.dot(y).item())
        y = y - r2 * (alfa / beta)
        r1 = r2
        r2 = y
        y = r2 if ident_M else Mop.matvec(r2)
        oldb = beta
        beta = float(r2.dot(y).item())
        if beta < 0:
            raise ValueError("minres: M is not positive definite")
        beta = math.sqrt(beta)
        # QR update of the tridiagonal (plane rotations)
        oldeps = epsln
        delta = cs * dbar + sn * alfa
        gbar = sn * dbar - cs * alfa
        epsln = sn * beta
        dbar = -cs * beta
        gamma = math.sqrt(gbar * gbar + beta * beta) or np.finfo(float).eps
        cs = gbar / gamma
        sn = beta / gamma
        phi = cs * phibar
        phibar = sn * phibar
        # update x
        w1 = w2
        w2 = w
        w = (v - w1 * oldeps - w2 * delta) * (1.0 / gamma)
        x += w * phi
        if callback is not None:
            callback(x)
        if abs(phibar) < threshold:
            info = 0
            break
    if info != 0:
        r = b - A.matvec(x)
        if float(r.norm().item()) < threshold:
            info = 0
    return x, info


# -- GMRES --------------------------------------------------------------------
@track_provenance(nested=True)
def gmres(A, b, x0=None, tol=1e-5, restart=None, maxiter=None, M=None,
          callback=None, atol=None):
    """Restarted GMRES with host-side least squares (reference
    linalg.py:670-792)."""
    A = aslinearoperator(A)
    b = _vec(b, A.dtype)
    n = b.shape[0]
    if restart is None:
        restart = min(20, n)
    if maxiter is None:
        maxiter = min(n * 10, 1000)
    M = aslinearoperator(M) if M is not None else None
    x = _vec(x0, A.dtype).copy() if x0 is not None else darray.zeros((n,), dtype=A.dtype)
    bnorm = float(b.norm().item()) or 1.0
    threshold = _tols(bnorm, tol, atol)
    cdtype = np.promote_types(A.dtype, np.float64)
    iters = 0
    while iters < maxiter:
        r = b - A.matvec(x)
        if M is not None:
            r = M.matvec(r)
        beta = float(r.norm().item())
        if beta < threshold:
            return x, 0
        # device-resident basis + batched CGS2 orthogonalization: 3
        # collectives + 1 host sync per Arnoldi step regardless of j
        # (was j+2 round-trips with modified Gram-Schmidt)
        Vb = _Basis(restart + 1, r)
        Vb.append(r.local / beta)
        H = np.zeros((restart + 1, restart), dtype=cdtype)
        j = 0
        while j < restart and iters < maxiter:
            w = A.matvec(Vb.row(j))
            if M is not None:
                w = M.matvec(w)
            wl = w.local.clone() if w.local.data_ptr() == Vb.V[j].data_ptr() \
                else w.local
            h1 = Vb.project(wl, j + 1)
            Vb.deflate(wl, h1, j + 1)
            h2 = Vb.project(wl, j + 1)  # CGS2: second pass restores MGS-level
            Vb.deflate(wl, h2, j + 1)   # orthogonality with batched reductions
            hcol = (h1 + h2).cpu().numpy()
            hh = Vb.global_norm(wl)
            H[: j + 1, j] = hcol
            H[j + 1, j] = hh
            if hh > 0:
                Vb.append(wl / hh)
            j += 1
            iters += 1
            # solve small lstsq on host, check residual
            e1 = np.zeros(j + 1, dtype=cdtype)
            e1[0] = beta
            y_h, res_, _, _ = np.linalg.lstsq(H[: j + 1, : j], e1, rcond=None)
            resid = float(np.linalg.norm(H[: j + 1, : j] @ y_h - e1))
            if resid < threshold or hh == 0:
                break
        y_dev = torch.as_tensor(np.asarray(y_h), device=Vb.V.device)
        x = x + DistArray.from_local(Vb.combine(y_dev, j), x.partition, x.shape)
        if callback is not None:
            callback(x)
        r = b - A.matvec(x)
        if M is not None:
            r = M.matvec(r)
        if float(r.norm().item()) < threshold:
            return x, 0
    return x, maxiter


# -- LSQR ---------------------------------------------------------------------
@track_provenance(nested=True)
def lsqr(A, b, damp=0.0, atol=1e-8, btol=1e-8, conlim=1e8, iter_lim=None,
         show=False, calc_var=False, x0=None):
    """Golub-Kahan LSQR (reference linalg.py:937-1415, itself scipy-lifted).
    Returns the scipy 10-tuple.

    The bidiagonalization consumes alfa/beta on the host each step (each
    normalizes the vector the NEXT matvec consumes), so ~2 syncs per
    iteration are inherent to the algorithm — same as scipy and the
    reference; the batched-reduction discipline applies to cg/gmres/eigsh
    where the recurrences permit it."""
    Aop = aslinearoperator(A)
    b = _vec(b, np.promote_types(Aop.dtype, np.float64))
    m, n = Aop.shape
    if iter_lim is None:
        iter_lim = 2 * n
    var = darray.zeros((n,), dtype=b.dtype)
    itn = 0
    istop = 0
    ctol = 1.0 / conlim if conlim > 0 else 0.0
    anorm = acond = 0.0
    dampsq = damp * damp
    ddnorm = res2 = xnorm = xxnorm = z = sn2 = 0.0
    cs2 = -1.0

    u = b.copy()
    bnorm = float(b.norm().item())
    if x0 is None:
        x = darray.zeros((n,), dtype=b.dtype)
        beta = bnorm
    else:
        x = _vec(x0, b.dtype).copy()
        u = u - Aop.matvec(x)
        beta = float(u.norm().item())
    if beta > 0:
        u = u * (1.0 / beta)
        v = Aop.rmatvec(u)
        alfa = float(v.norm().item())
    else:
        v = x.copy()
        alfa = 0.0
    if alfa > 0:
        v = v * (1.0 / alfa)
    w = v.copy()
    rhobar = alfa
    phibar = beta
    rnorm = r1norm = r2norm = beta
    arnorm = alfa * beta
    if arnorm == 0:
        return (x, 0, 0, r1norm, r2norm, anorm, acond, arnorm, xnorm, var)
    while itn < iter_lim:
        itn += 1
        u = Aop.matvec(v) - u * alfa
        beta = float(u.norm().item())
        if beta > 0:
            u = u * (1.0 / beta)
            anorm = math.sqrt(anorm**2 + alfa**2 + beta**2 + dampsq)
            v = Aop.rmatvec(u) - v * beta
            alfa = float(v.norm().item())
            if alfa > 0:
                v = v * (1.0 / alfa)
        if damp > 0:
            rhobar1 = math.sqrt(rhobar**2 + dampsq)
            cs1 = rhobar / rhobar1
            sn1 = damp / rhobar1
            psi = sn1 * phibar
            phibar = cs1 * phibar
        else:
            rhobar1 = rhobar
            psi = 0.0
        rho = math.sqrt(rhobar1**2 + beta**2)
        cs = rhobar1 / rho
        sn = beta / rho
        theta = sn * alfa
        rhobar = -cs * alfa
        phi = cs * phibar
        phibar = sn * phibar
        tau = sn * phi
        t1 = phi / rho
        t2 = -theta / rho
        dk = w * (1.0 / rho)
        x += w * t1
        w = v + w * t2
        ddnorm += float(dk.norm().item()) ** 2
        if calc_var:
            var += dk * dk
        delta = sn2 * rho
        gambar = -cs2 * rho
        rhs = phi - delta * z
        zbar = rhs / gambar if gambar != 0 else 0.0
        xnorm = math.sqrt(xxnorm + zbar**2)
        gamma = math.sqrt(gambar**2 + theta**2)
        if gamma != 0:
            cs2 = gambar / gamma
            sn2 = theta / gamma
            z = rhs / gamma
            xxnorm += z * z
        acond = anorm * math.sqrt(ddnorm)
        res1 = phibar**2
        res2 += psi**2
        rnorm = math.sqrt(res1 + res2)
        arnorm = alfa * abs(tau)
        r1sq = rnorm**2 - dampsq * xxnorm
        r1norm = math.sqrt(abs(r1sq))
        if r1sq < 0:
            r1norm = -r1norm
        r2norm = rnorm
        test1 = rnorm / bnorm if bnorm else np.inf
        test2 = arnorm / (anorm * rnorm) if anorm * rnorm else np.inf
        test3 = 1.0 / acond if acond else np.inf
        t1c = test1 / (1 + anorm * xnorm / bnorm) if bnorm else np.inf
        rtol_ = btol + atol * anorm * xnorm / bnorm if bnorm else 0.0
        if itn >= iter_lim:
            istop = 7
        if 1 + test3 <= 1:
            istop = 6
        if 1 + test2 <= 1:
            istop = 5
        if 1 + t1c <= 1:
            istop = 4
        if test3 <= ctol:
            istop = 3
        if test2 <= atol:
            istop = 2
        if test1 <= rtol_:
            istop = 1
        if istop != 0:
            break
    return (x, istop, itn, r1norm, r2norm, anorm, acond, arnorm, xnorm, var)


# -- eigsh (thick-restart Lanczos) -------------------------------------------
@track_provenance(nested=True)
def eigsh(a, k=6, which="LM", ncv=None, maxiter=None, tol=0.0):
    """Eigenpairs of a symmetric operator via thick-restart Lanczos
    (reference linalg.py:1416-1569, CuPy-lifted).  After a restart the
    projected matrix is an arrowhead (Ritz values on the diagonal, residual
    couplings into the new Lanczos vector) + the new tridiagonal block."""
    A = aslinearoperator(a)
    n = A.shape[0]
    if which not in ("LM", "LA", "SA"):
        raise ValueError(f"which={which} not supported")
    if k <= 0 or k >= n:
        raise ValueError("k must be in (0, n)")
    if ncv is None:
        ncv = min(max(2 * k, k + 32), n - 1)
    if maxiter is None:
        maxiter = 10 * n
    if tol == 0:
        tol = np.sqrt(np.finfo(np.float64).eps)
    if ncv <= k:
        # k == n-1 (tiny n): the Lanczos basis has no room past k —
        # materialize the operator by n matvecs and solve densely
        cols = []
        for i in range(n):
            e = np.zeros(n)
            e[i] = 1.0
            cols.append(np.asarray(A.matvec(e)))
        dense = np.stack(cols, axis=1)
        w_all, V_all = np.linalg.eigh(dense)
        if which == "LM":
            idx = np.argsort(np.abs(w_all))[::-1][:k]
        elif which == "LA":
            idx = np.argsort(w_all)[::-1][:k]
        else:
            idx = np.argsort(w_all)[:k]
        order = np.argsort(w_all[idx])
        return w_all[idx][order], V_all[:, idx][:, order]

    alpha = np.zeros(ncv)
    beta = np.zeros(ncv)
    u = darray.random((n,), dtype=np.float64, seed=7)
    u = u * (1.0 / float(u.norm().item()))
    # device-resident Lanczos basis: batched two-pass reorthogonalization,
    # 2 collectives + 2 host syncs per step regardless of j
    V = _Basis(ncv, u)
    V.append(u.local)
    restarted = False
    bcoup = np.zeros(k)

    def _select(w):
        if which == "LM":
            return np.argsort(np.abs(w))[::-1][:k]
        if which == "LA":
            return np.argsort(w)[::-1][:k]
        return np.argsort(w)[:k]

    def lanczos(start):
        nonlocal u
        for j in range(start, ncv):
            u = A.matvec(V.row(j))
            ul = u.local.clone() if u.local.data_ptr() == V.V[j].data_ptr() \
                else u.local
            c1 = V.project(ul, j + 1)  # first-pass coefficients: c1[j] = α_j
            alpha[j] = float(c1[j].item())
            V.deflate(ul, c1, j + 1)
            c2 = V.project(ul, j + 1)  # second pass: full reorthogonalization
            V.deflate(ul, c2, j + 1)
            b = V.global_norm(ul)
            beta[j] = b
            if b == 0:
                u = DistArray.from_local(ul, V.part, V.vshape)
                return
            ul = ul * (1.0 / b)
            u = DistArray.from_local(ul, V.part, V.vshape)
            if j + 1 < ncv:
                V.n = j + 1
                V.append(ul)

    lanczos(0)
    iters = ncv
    while True:
        T = np.diag(alpha)
        if restarted:
            T[:k, k] = bcoup
            T[k, :k] = bcoup
            for j in range(k, ncv - 1):
                T[j, j + 1] = T[j + 1, j] = beta[j]
        else:
            for j in range(ncv - 1):
                T[j, j + 1] = T[j + 1, j] = beta[j]
        w, s = np.linalg.eigh(T)
        idx = _select(w)
        wk, sk = w[idx], s[:, idx]
        res = np.abs(beta[ncv - 1] * sk[ncv - 1, :])
        if iters >= maxiter or np.all(res <= tol * np.maximum(1.0, np.abs(wk))):
            break
        # thick restart: V[:k] <- Ritz vectors, V[k] <- last Lanczos
        # residual — one batched (k x ncv) @ (ncv x nloc) matmul
        skd = torch.as_tensor(np.ascontiguousarray(sk.T), dtype=V.V.dtype,
                              device=V.V.device)
        V.V[:k] = skd @ V.V[:ncv]
        V.V[k].copy_(u.local)
        V.n = k + 1
        alpha[:k] = wk
        bcoup = beta[ncv - 1] * sk[ncv - 1, :]
        # one Lanczos step from V[k] against the arrowhead couplings
        unew = A.matvec(V.row(k))
        ul = unew.local.clone() if unew.local.data_ptr() == V.V[k].data_ptr() \
            else unew.local
        c1 = V.project(ul, k + 1)
        alpha[k] = float(c1[k].item())
        V.deflate(ul, c1, k + 1)
        c2 = V.project(ul, k + 1)
        V.deflate(ul, c2, k + 1)
        bnw = V.global_norm(ul)
        beta[k] = bnw
        if bnw == 0:
            restarted = True
            iters += 1
            continue
        ul = ul * (1.0 / bnw)
        u = DistArray.from_local(ul, V.part, V.vshape)
        if k + 1 < ncv:
            V.append(ul)
        lanczos(k + 1)
        restarted = True
        iters += ncv - k
    order = np.argsort(wk)
    wk = wk[order]
    sk = sk[:, order]
    skd = torch.as_tensor(np.ascontiguousarray(sk.T), dtype=V.V.dtype,
                          device=V.V.device)
    ritz = skd @ V.V[:ncv]  # (k, nloc)
    X = np.stack([np.asarray(DistArray.from_local(ritz[c], V.part, V.vshape))
                  for c in range(k)], axis=1)
    return wk, X
