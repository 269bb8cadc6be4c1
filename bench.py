"""Flagship benchmark: CG on the 5-pt 2-D Poisson operator (BASELINE.json).

python bench.py --gpus N --steps K --warmup W
  - one process per GPU (torchrun for N>1; RANK/WORLD_SIZE from env)
  - a "step" = one CG iteration (SpMV + 2 dots + 3 fused axpby) at
    n = nx^2 with nx=16384 (strong scaling across N GPUs; --weak scales
    nx by sqrt(N), the reference protocol)
  - rank 0 prints ONE JSON line; value = whole-job CG iterations/sec.

Besides the headline it measures (a) standalone CSR SpMV GFLOP/s and
(b) the MATCHED reference config — nx=6000 per GPU, sqrt(N)-weak-scaled,
the exact grid the reference's 75.9 it/s V100 number is quoted on
(results/summit/legate_gpu_pde.out) — so both the headline-vs-baseline
and the apples-to-apples ratios are machine-readable (config.matched_*).
"""
from __future__ import annotations

import argparse
import json
import math
import os
import sys
import time

import numpy as np
import torch

BASELINE_ITERS_PER_SEC = 75.9  # reference, 1 V100, nx=6000
MATCHED_NX = 6000


def make_cg_stepper(A, b, n, dtype, on_gpu, allow_graph):
    """Build a no-convergence-check CG stepper over (A, b); returns
    (cg_step, state) where state keeps buffers alive (graph safety)."""
    import sparse  # noqa: F401
    from sparse import darray, linalg
    from sparse.parallel import comm

    Aop = linalg.aslinearoperator(A)
    xv = darray.zeros((n,), dtype=dtype)
    r = b - Aop.matvec(xv)
    pvec = r.copy()
    rz = r.dot(r)
    q = darray.zeros((n,), dtype=dtype)
    state = {"x": xv, "r": r, "p": pvec, "q": q, "rz": rz}

    if not on_gpu:
        def cg_step() -> None:
            Aop.matvec(pvec, out=q)
            pq = pvec.dot(q)
            linalg.cg_axpby(xv, pvec, state["rz"], pq, isalpha=True, negate=False)
            linalg.cg_axpby(r, q, state["rz"], pq, isalpha=True, negate=True)
            rz_new = r.dot(r)
            linalg.cg_axpby(pvec, r, rz_new, state["rz"], isalpha=False,
                            negate=False)
            state["rz"] = rz_new

        return cg_step, state

    from sparse import kernels

    rz_buf = rz.clone()
    state["rz_buf"] = rz_buf
    use2 = A._dia() is not None and os.environ.get("SPARSE_CG2") == "1"
    if use2:
        # two-kernel CG iteration (opt-in: measured slower than the
        # 4-kernel loop on MI355X — see sparse/linalg.py cg())
        rz_old_buf = rz.clone()
        p_b = darray.zeros((n,), dtype=dtype)
        state.update({"rz_old": rz_old_buf, "p_b": p_b})

        def prologue() -> None:  # iteration 0: p is r, beta undefined
            pq = A.spmv_dot(pvec, q)
            rz_new = kernels.cg_xr_norm2(xv.local, pvec.local, r.local,
                                         q.local, rz_buf, pq)
            comm.all_reduce_(rz_new)
            rz_old_buf.copy_(rz_buf)
            rz_buf.copy_(rz_new)

        def step2(pc, pn) -> None:
            pq = A.spmv_bpdot(r, pc, pn, q, rz_buf, rz_old_buf)
            rz_new = kernels.cg_xr_norm2(xv.local, pn.local, r.local,
                                         q.local, rz_buf, pq)
            comm.all_reduce_(rz_new)
            rz_old_buf.copy_(rz_buf)
            rz_buf.copy_(rz_new)

        prologue()
        _parity = [0]

        def cg_step_gpu() -> None:
            if _parity[0] == 0:
                step2(pvec, p_b)
            else:
                step2(p_b, pvec)
            _parity[0] ^= 1

    else:

        use3 = not os.environ.get("SPARSE_CG4")

        def cg_step_gpu() -> None:  # fused 3-kernel path (see linalg.cg)
            # rz_buf is a persistent 0-dim buffer so the step is
            # hipGraph-capturable (fixed addresses across replays)
            pq = A.spmv_dot(pvec, q)
            if use3:
                # K2 fuses x += (rz/pq)p and r -= (rz/pq)q + |r|^2: same
                # HBM bytes as the two axpbys it replaces, one kernel less
                rz_new = kernels.cg_xr_norm2(xv.local, pvec.local, r.local,
                                             q.local, rz_buf, pq)
                comm.all_reduce_(rz_new)
            else:  # SPARSE_CG4=1: the round-1 4-kernel formulation
                linalg.cg_axpby(xv, pvec, rz_buf, pq, isalpha=True,
                                negate=False)
                rz_new = linalg._axpby_norm2(r, q, rz_buf, pq, negate=True)
            linalg.cg_axpby(pvec, r, rz_new, rz_buf, isalpha=False, negate=False)
            rz_buf.copy_(rz_new)

    cg_step = cg_step_gpu
    if allow_graph and not os.environ.get("SPARSE_NO_HIPGRAPH"):
        # capture CG iterations as hipGraphs (launch-overhead free replay);
        # multi-GPU stays eager here (RCCL collectives outside capture).
        # The two-kernel path double-buffers p: capture BOTH parities.
        try:
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                for _ in range(4):
                    cg_step_gpu()
            torch.cuda.current_stream().wait_stream(side)
            if use2:
                _parity[0] = 0
                g_ab = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g_ab):
                    step2(pvec, p_b)
                g_ba = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g_ba):
                    step2(p_b, pvec)
                state["graph"] = (g_ab, g_ba)

                def cg_step() -> None:
                    state["graph"][_parity[0]].replay()
                    _parity[0] ^= 1

            else:
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g):
                    cg_step_gpu()
                state["graph"] = g
                cg_step = g.replay
        except Exception as e:
            print(f"# hipGraph capture unavailable ({e}); eager steps",
                  file=sys.stderr)
    return cg_step, state


def time_steps(cg_step, steps, warmup, sync):
    from sparse.parallel import comm

    for _ in range(warmup):
        cg_step()
    sync()
    t0 = time.time()
    for _ in range(steps):
        cg_step()
    sync()
    elapsed = time.time() - t0
    et = torch.tensor([elapsed])
    comm.all_reduce_(et, op="max")  # max over ranks
    return float(et.item())


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=300)
    p.add_argument("--warmup", type=int, default=20)
    p.add_argument("--nx", type=int, default=16384)
    p.add_argument("--dtype", default="fp64", choices=["fp64", "fp32"])
    p.add_argument("--weak", action="store_true",
                   help="scale nx by sqrt(world) (reference weak-scaling protocol)")
    p.add_argument("--no-matched", action="store_true",
                   help="skip the matched-config (nx=6000/GPU) measurement")
    args = p.parse_args()

    import sparse  # noqa: F401
    from sparse import darray, gallery
    from sparse.parallel import comm
    from sparse.runtime import runtime

    rt = runtime()
    on_gpu = rt.use_gpu
    if on_gpu:
        from sparse.kernels import require

        require()
    dtype = np.float64 if args.dtype == "fp64" else np.float32
    ws = max(1, comm.world_size())

    nx = args.nx
    if args.weak:
        nx = int(round(args.nx * math.sqrt(ws)))
    n = nx * nx
    t0 = time.time()
    A = gallery.poisson2d(nx, dtype=dtype)
    b = darray.ones((n,), dtype=dtype)
    build_s = time.time() - t0

    def sync() -> None:
        if on_gpu:
            torch.cuda.synchronize()
        comm.initialized() and torch.distributed.barrier()
        if on_gpu:
            torch.cuda.synchronize()

    # -- standalone SpMV measurement (secondary metric) ----------------------
    x = darray.ones((n,), dtype=dtype)
    y = darray.zeros((n,), dtype=dtype)
    for _ in range(3):
        A.dot(x, out=y)
    sync()
    t0 = time.time()
    SPMV_IT = 10
    for _ in range(SPMV_IT):
        A.dot(x, out=y)
    sync()
    spmv_s = (time.time() - t0) / SPMV_IT
    nnz = A.nnz
    spmv_gflops = 2.0 * nnz / spmv_s / 1e9
    # bytes: vals + idx + indptr + x-read + y-write (first-order)
    idx_b = 4 if max(A.shape) < 2**31 - 1 else 8
    vb = np.dtype(dtype).itemsize
    spmv_gbps = (nnz * (vb + idx_b) + n * (3 * vb)) / spmv_s / 1e9
    del x, y

    # -- headline CG loop ----------------------------------------------------
    # ws>1 capture (RCCL collectives inside the hipGraph) is opt-in until
    # validated on a multi-GPU box: SPARSE_WS_HIPGRAPH=1; a failed capture
    # falls back to eager inside make_cg_stepper
    allow_graph = on_gpu and (
        ws == 1 or os.environ.get("SPARSE_WS_HIPGRAPH") == "1")
    cg_step, _state = make_cg_stepper(A, b, n, dtype, on_gpu, allow_graph)
    elapsed = time_steps(cg_step, args.steps, args.warmup, sync)
    iters_per_sec = args.steps / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    # -- matched reference config: nx=6000 per GPU, sqrt(N)-weak ------------
    matched = None
    m_nx = int(round(MATCHED_NX * math.sqrt(ws)))
    if on_gpu and not args.no_matched and m_nx != nx:
        del cg_step, _state, A, b
        torch.cuda.empty_cache()
        m_n = m_nx * m_nx
        A2 = gallery.poisson2d(m_nx, dtype=dtype)
        b2 = darray.ones((m_n,), dtype=dtype)
        step2, _state2 = make_cg_stepper(A2, b2, m_n, dtype, on_gpu, allow_graph)
        m_steps = max(200, args.steps)
        m_elapsed = time_steps(step2, m_steps, 25, sync)
        matched = {
            "matched_nx": m_nx,
            "matched_steps": m_steps,
            "matched_iters_per_sec": round(m_steps / m_elapsed, 2),
            "matched_vs_baseline": round(
                m_steps / m_elapsed / BASELINE_ITERS_PER_SEC, 3),
        }

    if comm.rank() == 0:
        out = {
            "metric": "cg_iters_per_sec",
            "value": iters_per_sec,
            "unit": "iters/s",
            "n_gpus": args.gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak" if args.weak else "strong",
            "vs_baseline": iters_per_sec / BASELINE_ITERS_PER_SEC,
            "dtype": args.dtype,
            "data": "synthetic",
            "config": {
                "model": "poisson2d-5pt-cg",
                "nx": nx,
                "n": n,
                "nnz": nnz,
                "parallelism": f"dp{args.gpus} 1-D row partition + window gather",
                "spmv_gflops": round(spmv_gflops, 2),
                "spmv_effective_gbps": round(spmv_gbps, 1),
                "build_s": round(build_s, 2),
                **(matched or {}),
            },
        }
        print(json.dumps(out))


if __name__ == "__main__":
    main()
