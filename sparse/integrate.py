"""ODE integration: solve_ivp with RK23 / RK45 / DOP853 on DistArrays.

Reference parity: sparse/integrate.py (scipy-lifted solve_ivp with a
RungeKutta base integrate.py:619-747, RK23 :750, RK45 :838, DOP853 :987,
dense outputs, event handling :1175-1301) and the fused RK stage
combination task RK_CALC_DY (integrate.py:478-494, runge_kutta.cu:26-42) —
here kernels.rk_calc_dy on GPU, a torch matvec on CPU.

State vectors are DistArrays (row-partitioned); stage matrices K are local
(s, n_local) torch tensors; error norms are global RMS via one all-reduce.
"""
from __future__ import annotations

import math
import numpy as np
import torch

from . import dop853_coefficients as dop853
from .darray import DistArray, asdistarray
from .parallel import comm

__all__ = ["solve_ivp", "RK23", "RK45", "DOP853", "OdeSolution"]

SAFETY = 0.9
MIN_FACTOR = 0.2
MAX_FACTOR = 10.0


def _rms_norm(local_sq_sum: torch.Tensor, n: int) -> float:
    t = local_sq_sum.clone()
    comm.all_reduce_(t)
    return math.sqrt(max(float(t.item()), 0.0) / n)


def _sq(t: torch.Tensor) -> torch.Tensor:
    return torch.sum(torch.abs(t) ** 2) if t.is_complex() else torch.sum(t * t)


def _err_norm(e_local: torch.Tensor, scale_local: torch.Tensor, n: int) -> float:
    return _rms_norm(_sq(e_local / scale_local), n)


def _combine(K: torch.Tensor, coeffs: np.ndarray, h: float) -> torch.Tensor:
    """dy_local = h * sum_j K[j] * coeffs[j] — the fused RK_CALC_DY."""
    c = torch.as_tensor(coeffs, dtype=K.dtype, device=K.device)
    if K.is_cuda:
        from . import kernels

        kernels.require()
        dy = torch.empty(K.shape[1], dtype=K.dtype, device=K.device)
        kernels.rk_calc_dy(K.contiguous(), c.contiguous(), float(h), dy)
        return dy
    return h * (c @ K)


class RungeKutta:
    """Adaptive explicit RK (scipy architecture; reference
    integrate.py:619-747)."""

    C: np.ndarray
    A: np.ndarray
    B: np.ndarray
    E: np.ndarray
    order: int
    error_estimator_order: int
    n_stages: int

    def __init__(self, fun, t0, y0: DistArray, t_bound, rtol=1e-3, atol=1e-6,
                 max_step=np.inf, first_step=None):
        self.fun = fun
        self.t = float(t0)
        self.y = y0.copy()
        self.t_bound = float(t_bound)
        self.direction = 1.0 if t_bound >= t0 else -1.0
        self.rtol, self.atol = rtol, atol
        self.max_step = max_step
        self.n = y0.shape[0]
        self.f = asdistarray(fun(self.t, self.y))
        self.status = "running"
        self.t_old = None
        self.y_old = None
        self.K = torch.empty((self.n_stages + 1, self.y.local.shape[0]),
                             dtype=self.y.local.dtype, device=self.y.local.device)
        self.error_exponent = -1.0 / (self.error_estimator_order + 1)
        if first_step is None:
            self.h_abs = self._select_initial_step()
        else:
            self.h_abs = float(first_step)

    def _scale_local(self, y_l, yn_l=None):
        m = torch.abs(y_l) if yn_l is None else torch.maximum(
            torch.abs(y_l), torch.abs(yn_l))
        return self.atol + self.rtol * m

    def _select_initial_step(self):
        """scipy's select_initial_step (reference integrate.py:127)."""
        y, f = self.y, self.f
        scale = self._scale_local(y.local)
        d0 = _rms_norm(_sq(y.local / scale), self.n)
        d1 = _rms_norm(_sq(f.local / scale), self.n)
        h0 = 1e-6 if d0 < 1e-5 or d1 < 1e-5 else 0.01 * d0 / d1
        y1 = y + f * (h0 * self.direction)
        f1 = asdistarray(self.fun(self.t + h0 * self.direction, y1))
        d2 = _rms_norm(_sq((f1.local - f.local) / scale), self.n) / h0
        if d1 <= 1e-15 and d2 <= 1e-15:
            h1 = max(1e-6, h0 * 1e-3)
        else:
            h1 = (0.01 / max(d1, d2)) ** (1.0 / (self.order + 1))
        return min(100 * h0, h1, self.max_step,
                   abs(self.t_bound - self.t) or np.inf)

    def _estimate_error_local(self, h):
        return _combine(self.K, self.E, h)

    def step(self):
        if self.status != "running":
            raise RuntimeError("attempt to step on a finished solver")
        t = self.t
        max_step = self.max_step
        min_step = 10 * abs(np.nextafter(t, self.direction * np.inf) - t)
        h_abs = min(max(self.h_abs, min_step), max_step)
        step_accepted = False
        step_rejected = False
        while not step_accepted:
            if h_abs < min_step:
                self.status = "failed"
                return False
            h = h_abs * self.direction
            t_new = t + h
            if self.direction * (t_new - self.t_bound) > 0:
                t_new = self.t_bound
            h = t_new - t
            h_abs = abs(h)

            y_new_l, f_new = self._rk_step(h)
            scale = self._scale_local(self.y.local, y_new_l)
            err = _err_norm(self._estimate_error_local(h), scale, self.n)
            if err < 1.0:
                factor = MAX_FACTOR if err == 0 else min(
                    MAX_FACTOR, SAFETY * err ** self.error_exponent)
                if step_rejected:
                    factor = min(1.0, factor)
                h_abs *= factor
                step_accepted = True
            else:
                h_abs *= max(MIN_FACTOR, SAFETY * err ** self.error_exponent)
                step_rejected = True
        self.h_previous = h
        self.t_old = t
        self.y_old = self.y
        self.t = t_new
        self.y = DistArray.from_local(y_new_l, self.y.partition, self.y.gshape)
        self.f = f_new
        self.h_abs = h_abs
        if self.direction * (self.t - self.t_bound) >= 0:
            self.status = "finished"
        return True

    def _rk_step(self, h):
        """Stage evaluations + fused combinations (reference rk_step,
        integrate.py:498)."""
        t, y = self.t, self.y
        self.K[0] = self.f.local
        for s in range(1, self.n_stages):
            dy_l = _combine(self.K[:s], self.A[s, :s], h)
            ys = DistArray.from_local(y.local + dy_l, y.partition, y.gshape)
            self.K[s] = asdistarray(self.fun(t + self.C[s] * h, ys)).local
        y_new_l = y.local + _combine(self.K[: self.n_stages], self.B, h)
        f_new = asdistarray(self.fun(t + h, DistArray.from_local(
            y_new_l, y.partition, y.gshape)))
        self.K[self.n_stages] = f_new.local
        return y_new_l, f_new

    def dense_output(self):
        Q_l = self.K[: self.n_stages + 1].T @ torch.as_tensor(
            self.P, dtype=self.K.dtype, device=self.K.device)
        return RkDenseOutput(self.t_old, self.t, self.y_old, Q_l)


class RK23(RungeKutta):
    """Bogacki-Shampine 3(2) (reference integrate.py:750)."""

    order = 3
    error_estimator_order = 2
    n_stages = 3
    C = np.array([0, 1 / 2, 3 / 4])
    A = np.array([[0, 0, 0], [1 / 2, 0, 0], [0, 3 / 4, 0]])
    B = np.array([2 / 9, 1 / 3, 4 / 9])
    E = np.array([5 / 72, -1 / 12, -1 / 9, 1 / 8])
    P = np.array([[1, -4 / 3, 5 / 9], [0, 1, -2 / 3], [0, 4 / 3, -8 / 9],
                  [0, -1, 1]])


class RK45(RungeKutta):
    """Dormand-Prince 5(4) (reference integrate.py:838)."""

    order = 5
    error_estimator_order = 4
    n_stages = 6
    C = np.array([0, 1 / 5, 3 / 10, 4 / 5, 8 / 9, 1])
    A = np.array([
        [0, 0, 0, 0, 0],
        [1 / 5, 0, 0, 0, 0],
        [3 / 40, 9 / 40, 0, 0, 0],
        [44 / 45, -56 / 15, 32 / 9, 0, 0],
        [19372 / 6561, -25360 / 2187, 64448 / 6561, -212 / 729, 0],
        [9017 / 3168, -355 / 33, 46732 / 5247, 49 / 176, -5103 / 18656],
    ])
    B = np.array([35 / 384, 0, 500 / 1113, 125 / 192, -2187 / 6784, 11 / 84])
    E = np.array([71 / 57600, 0, -71 / 16695, 71 / 1920, -17253 / 339200,
                  22 / 525, -1 / 40])
    P = np.array([
        [1, -8048581381 / 2820520608, 8663915743 / 2820520608,
         -12715105075 / 11282082432],
        [0, 0, 0, 0],
        [0, 131558114200 / 32700410799, -68118460800 / 10900136933,
         87487479700 / 32700410799],
        [0, -1754552775 / 470086768, 14199869525 / 1410260304,
         -10690763975 / 1880347072],
        [0, 127303824393 / 49829197408, -318862633887 / 49829197408,
         701980252875 / 199316789632],
        [0, -282668133 / 205662961, 2019193451 / 616988883,
         -1453857185 / 822651844],
        [0, 40617522 / 29380423, -110615467 / 29380423, 69997945 / 29380423],
    ])


class DOP853(RungeKutta):
    """Hairer's 8(5,3) (reference integrate.py:987)."""

    order = 8
    error_estimator_order = 7
    n_stages = dop853.N_STAGES
    C = dop853.C[: dop853.N_STAGES]
    A = dop853.A[: dop853.N_STAGES, : dop853.N_STAGES]
    B = dop853.B
    E3 = dop853.E3
    E5 = dop853.E5
    D = dop853.D
    A_EXTRA = dop853.A[dop853.N_STAGES + 1:]
    C_EXTRA = dop853.C[dop853.N_STAGES + 1:]

    def __init__(self, *a, **kw):
        super().__init__(*a, **kw)
        self.K_extended = torch.empty(
            (dop853.N_STAGES_EXTENDED, self.y.local.shape[0]),
            dtype=self.y.local.dtype, device=self.y.local.device)
        self.K = self.K_extended[: self.n_stages + 1]

    def _estimate_error_local(self, h):
        # the 8(5,3) double error estimate (scipy's formulation)
        err5 = _combine(self.K, self.E5, 1.0)
        err3 = _combine(self.K, self.E3, 1.0)
        return (err5, err3, h)

    def step(self):  # override error handling via custom norm
        return super().step()


def _dop853_err_norm(self_obj, est, scale, n):
    err5, err3, h = est
    s5 = _sq(err5 / scale)
    s3 = _sq(err3 / scale)
    t = torch.stack([s5, s3])
    comm.all_reduce_(t)
    # scipy's 8(5,3) norm: |h| * s5 / sqrt((s5 + 0.01*s3) * n) on the
    # UN-normalized global squared sums (reference integrate.py:1117-1130);
    # dividing by n first made step control ~sqrt(n) looser than rtol.
    s5 = float(t[0].item())
    s3 = float(t[1].item())
    denom = s5 + 0.01 * s3
    if denom <= 0:
        return 0.0
    return abs(h) * s5 / math.sqrt(denom * n)


# patch the norm dispatch: RungeKutta.step calls _err_norm on locals; DOP853
# returns a tuple — handle both.
_base_err_norm = _err_norm


def _dispatch_err_norm(est, scale, n, solver=None):
    if isinstance(est, tuple):
        return _dop853_err_norm(solver, est, scale, n)
    return _base_err_norm(est, scale, n)


# rebind inside RungeKutta.step via method override
_orig_step = RungeKutta.step


def _step_with_dispatch(self):
    if self.status != "running":
        raise RuntimeError("attempt to step on a finished solver")
    t = self.t
    min_step = 10 * abs(np.nextafter(t, self.direction * np.inf) - t)
    h_abs = min(max(self.h_abs, min_step), self.max_step)
    step_accepted = False
    step_rejected = False
    while not step_accepted:
        if h_abs < min_step:
            self.status = "failed"
            return False
        h = h_abs * self.direction
        t_new = t + h
        if self.direction * (t_new - self.t_bound) > 0:
            t_new = self.t_bound
        h = t_new - t
        h_abs = abs(h)
        y_new_l, f_new = self._rk_step(h)
        scale = self._scale_local(self.y.local, y_new_l)
        err = _dispatch_err_norm(self._estimate_error_local(h), scale, self.n,
                                 solver=self)
        if err < 1.0:
            factor = MAX_FACTOR if err == 0 else min(
                MAX_FACTOR, SAFETY * err ** self.error_exponent)
            if step_rejected:
                factor = min(1.0, factor)
            h_abs *= factor
            step_accepted = True
        else:
            h_abs *= max(MIN_FACTOR, SAFETY * err ** self.error_exponent)
            step_rejected = True
    self.h_previous = h
    self.t_old = t
    self.y_old = self.y
    self.t = t_new
    self.y = DistArray.from_local(y_new_l, self.y.partition, self.y.gshape)
    self.f = f_new
    self.h_abs = h_abs
    if self.direction * (self.t - self.t_bound) >= 0:
        self.status = "finished"
    return True


RungeKutta.step = _step_with_dispatch


def _dop853_dense(self):
    """Extended-stage dense output (scipy's _dense_output_impl)."""
    K = self.K_extended
    h = self.h_previous
    for s_i, (a, c) in enumerate(zip(self.A_EXTRA, self.C_EXTRA),
                                 start=dop853.N_STAGES + 1):
        dy_l = _combine(K[:s_i], a[:s_i], h)
        ys = DistArray.from_local(self.y_old.local + dy_l,
                                  self.y.partition, self.y.gshape)
        K[s_i] = asdistarray(self.fun(self.t_old + c * h, ys)).local
    F = torch.empty((dop853.INTERPOLATOR_POWER, K.shape[1]),
                    dtype=K.dtype, device=K.device)
    f_old = K[0]
    delta_y = self.y.local - self.y_old.local
    F[0] = delta_y
    F[1] = h * f_old - delta_y
    F[2] = 2 * delta_y - h * (self.f.local + f_old)
    Dm = torch.as_tensor(self.D, dtype=K.dtype, device=K.device)
    F[3:] = h * (Dm @ K)
    return Dop853DenseOutput(self.t_old, self.t, self.y_old, F)


DOP853.dense_output = _dop853_dense


class RkDenseOutput:
    def __init__(self, t_old, t, y_old: DistArray, Q_l: torch.Tensor):
        self.t_old, self.t = t_old, t
        self.h = t - t_old
        self.y_old = y_old
        self.Q = Q_l
        self.order = Q_l.shape[1] - 1

    def __call__(self, t):
        x = (t - self.t_old) / self.h
        p = np.cumprod(np.full(self.order + 1, x)) * self.h
        pt = torch.as_tensor(p, dtype=self.Q.dtype, device=self.Q.device)
        y_l = self.y_old.local + self.Q @ pt
        return DistArray.from_local(y_l, self.y_old.partition, self.y_old.gshape)


class Dop853DenseOutput:
    def __init__(self, t_old, t, y_old: DistArray, F: torch.Tensor):
        self.t_old, self.t = t_old, t
        self.h = t - t_old
        self.y_old = y_old
        self.F = F

    def __call__(self, t):
        x = float((t - self.t_old) / self.h)
        y_l = torch.zeros_like(self.y_old.local)
        for i, f in enumerate(reversed(self.F)):
            y_l += f
            y_l *= x if i % 2 == 0 else (1 - x)
        y_l += self.y_old.local
        return DistArray.from_local(y_l, self.y_old.partition, self.y_old.gshape)


class OdeSolution:
    def __init__(self, ts, interpolants):
        self.ts = np.asarray(ts)
        self.interpolants = interpolants

    def __call__(self, t):
        i = np.searchsorted(self.ts, t, side="left")
        i = int(np.clip(i - 1, 0, len(self.interpolants) - 1))
        return self.interpolants[i](t)


class OdeResult(dict):
    def __getattr__(self, k):
        try:
            return self[k]
        except KeyError as e:
            raise AttributeError(k) from e


METHODS = {"RK23": RK23, "RK45": RK45, "DOP853": DOP853}


def solve_ivp(fun, t_span, y0, method="RK45", t_eval=None, dense_output=False,
              events=None, rtol=1e-3, atol=1e-6, max_step=np.inf,
              first_step=None, args=None):
    """scipy-compatible solve_ivp on DistArrays (reference
    integrate.py:1303-...)."""
    if args is not None:
        _fun = fun
        fun = lambda t, y: _fun(t, y, *args)
    t0, tf = map(float, t_span)
    y0 = asdistarray(y0)
    if method not in METHODS:
        raise ValueError(f"method must be one of {list(METHODS)}")
    solver = METHODS[method](fun, t0, y0, tf, rtol=rtol, atol=atol,
                             max_step=max_step, first_step=first_step)

    if events is not None and not isinstance(events, (list, tuple)):
        events = [events]
    ev_vals = None
    if events:
        ev_vals = [float(e(t0, y0)) for e in events]
    t_events = [[] for _ in (events or [])]
    y_events = [[] for _ in (events or [])]

    ts = [t0]
    ys = [y0.copy()]
    interpolants = []
    status = None
    while solver.status == "running":
        ok = solver.step()
        if not ok:
            status = -1
            break
        sol = None
        if dense_output or t_eval is not None or events:
            sol = solver.dense_output()
            interpolants.append(sol)
        if events:
            new_vals = [float(e(solver.t, solver.y)) for e in events]
            for ei, (e, v0, v1) in enumerate(zip(events, ev_vals, new_vals)):
                direction = getattr(e, "direction", 0)
                crossed = (v0 < 0 <= v1 and direction >= 0) or (
                    v0 > 0 >= v1 and direction <= 0)
                if crossed and v0 != v1:
                    # bisection on the dense output
                    lo, hi = solver.t_old, solver.t
                    for _ in range(50):
                        mid = 0.5 * (lo + hi)
                        vm = float(e(mid, sol(mid)))
                        if (v0 < 0) == (vm < 0):
                            lo = mid
                        else:
                            hi = mid
                    te = 0.5 * (lo + hi)
                    t_events[ei].append(te)
                    y_events[ei].append(np.asarray(sol(te)))
                    if getattr(e, "terminal", False):
                        status = 1
                        solver.status = "finished"
            ev_vals = new_vals
        ts.append(solver.t)
        ys.append(solver.y.copy())
        if status == 1:
            break
    if status is None:
        status = 0 if solver.status == "finished" else -1

    ts_np = np.asarray(ts)
    if t_eval is not None:
        sol_obj = OdeSolution(ts_np, interpolants)
        t_out = np.asarray(t_eval, dtype=float)
        y_out = np.stack([np.asarray(sol_obj(t)) for t in t_out], axis=1)
        t_res = t_out
    else:
        t_res = ts_np
        y_out = np.stack([np.asarray(y) for y in ys], axis=1)
    return OdeResult(
        t=t_res, y=y_out, status=status, success=status >= 0,
        sol=OdeSolution(ts_np, interpolants) if dense_output else None,
        t_events=[np.asarray(te) for te in t_events] if events else None,
        y_events=y_events if events else None,
        nfev=-1, njev=0, nlu=0,
        message="done" if status >= 0 else "failed",
    )
