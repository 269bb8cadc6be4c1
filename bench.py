"""Flagship benchmark: CG on the 5-pt 2-D Poisson operator (BASELINE.json).

python bench.py --gpus N --steps K --warmup W
  - one process per GPU (torchrun for N>1; RANK/WORLD_SIZE from env)
  - a "step" = one CG iteration (SpMV + 2 dots + 3 fused axpby) at
    n = nx^2 with nx=16384 (strong scaling across N GPUs)
  - rank 0 prints ONE JSON line; value = whole-job CG iterations/sec.

Also measures standalone CSR SpMV GFLOP/s (reported inside config).
Reference baseline: 75.9 CG iters/s on 1 V100 (nx=6000).
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import numpy as np
import torch


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=60)
    p.add_argument("--warmup", type=int, default=15)
    p.add_argument("--nx", type=int, default=16384)
    p.add_argument("--dtype", default="fp64", choices=["fp64", "fp32"])
    p.add_argument("--weak", action="store_true",
                   help="scale nx by sqrt(world) (reference weak-scaling protocol)")
    args = p.parse_args()

    import sparse
    from sparse import darray, gallery, linalg
    from sparse.parallel import comm
    from sparse.runtime import runtime

    rt = runtime()
    on_gpu = rt.use_gpu
    if on_gpu:
        from sparse.kernels import require

        require()
    dtype = np.float64 if args.dtype == "fp64" else np.float32

    nx = args.nx
    if args.weak:
        import math

        from sparse.parallel import comm as _c

        nx = int(round(args.nx * math.sqrt(max(1, _c.world_size()))))
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

    # -- CG iteration loop, fixed step count (no convergence break) ----------
    Aop = linalg.aslinearoperator(A)
    xv = darray.zeros((n,), dtype=dtype)
    r = b - Aop.matvec(xv)
    pvec = r.copy()
    rz = r.dot(r)
    q = darray.zeros((n,), dtype=dtype)

    graph = None
    if on_gpu:
        from sparse import kernels

        rz_buf = rz.clone()
        # two-kernel CG iteration (opt-in: measured slower than the
        # 4-kernel loop on MI355X — see sparse/linalg.py cg())
        use2 = A._dia() is not None and os.environ.get("SPARSE_CG2") == "1"
        if use2:
            # K1 folds p = r + beta*p into the SpMV (q = Ap, p.q fused);
            # K2 fuses x += alpha p, r -= alpha q and |r|^2 — two HBM
            # passes per iteration fewer than the 4-kernel loop.
            rz_old_buf = rz.clone()
            p_b = darray.zeros((n,), dtype=dtype)

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

            def cg_step_gpu() -> None:  # fused 4-kernel path (see linalg.cg)
                # rz_buf is a persistent 0-dim buffer so the step is
                # hipGraph-capturable (fixed addresses across replays)
                pq = A.spmv_dot(pvec, q)
                linalg.cg_axpby(xv, pvec, rz_buf, pq, isalpha=True, negate=False)
                rz_new = linalg._axpby_norm2(r, q, rz_buf, pq, negate=True)
                linalg.cg_axpby(pvec, r, rz_new, rz_buf, isalpha=False, negate=False)
                rz_buf.copy_(rz_new)

        cg_step = cg_step_gpu
        if args.gpus == 1 and not os.environ.get("SPARSE_NO_HIPGRAPH"):
            # capture CG iterations as hipGraphs (launch-overhead free
            # replay); multi-GPU stays eager (RCCL outside graphs).  The
            # two-kernel path double-buffers p, so capture BOTH parities
            # and alternate replays.
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
                    graph = (g_ab, g_ba)

                    def cg_step() -> None:
                        graph[_parity[0]].replay()
                        _parity[0] ^= 1

                else:
                    g = torch.cuda.CUDAGraph()
                    with torch.cuda.graph(g):
                        cg_step_gpu()
                    graph = g
                    cg_step = graph.replay
            except Exception as e:
                print(f"# hipGraph capture unavailable ({e}); eager steps",
                      file=sys.stderr)

    else:

        def cg_step() -> None:
            nonlocal rz
            Aop.matvec(pvec, out=q)
            pq = pvec.dot(q)
            linalg.cg_axpby(xv, pvec, rz, pq, isalpha=True, negate=False)
            linalg.cg_axpby(r, q, rz, pq, isalpha=True, negate=True)
            rz_new = r.dot(r)
            linalg.cg_axpby(pvec, r, rz_new, rz, isalpha=False, negate=False)
            rz = rz_new

    for _ in range(args.warmup):
        cg_step()
    sync()
    t0 = time.time()
    for _ in range(args.steps):
        cg_step()
    sync()
    elapsed = time.time() - t0
    # max over ranks
    et = torch.tensor([elapsed])
    comm.all_reduce_(et, op="max")
    elapsed = float(et.item())
    iters_per_sec = args.steps / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

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
            "vs_baseline": iters_per_sec / 75.9,
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
            },
        }
        print(json.dumps(out))


if __name__ == "__main__":
    main()
