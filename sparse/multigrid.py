"""Replicated coarse-tail V-cycle for multigrid preconditioners.

Below a size threshold every rank holds the FULL coarse matrices and runs
the identical local V-cycle — zero collectives per coarse level instead of
a latency-bound halo exchange each, and on GPU the whole sub-cycle is one
hipGraph replay.  This is the MI355X realization of the reference's
machine-scoping (examples/gmg.py:212-218), extended from 'fewer ranks' to
'all ranks, replicated' because xGMI makes the one transition all-gather
cheap while eager launch latency is what killed the reference at scale
(BASELINE.md GMG 37.2 -> 1.5 it/s).

Used by examples/gmg.py (GMG) and examples/amg.py (smoothed-aggregation
AMG).  Smoothing in the replicated tail is weighted Jacobi.
"""
from __future__ import annotations

import os

import numpy as np
import torch


class ReplicatedCoarseCycle:
    """levels: objects with .A (distributed csr_array), .dinv (DistArray),
    .omega (float), and .Rdown/.Pdown transfer csr_arrays on every level
    except the last.  The last level may omit dinv/omega (dense solve).
    Construction is COLLECTIVE (gathers each level to every rank)."""

    def __init__(self, levels, coarse_inv_t: torch.Tensor, smooth_iters: int):
        from .ops import local as ops

        self.smooth_iters = smooth_iters
        self.coarse_inv_t = coarse_inv_t
        dev = coarse_inv_t.device
        self.on_gpu = dev.type == "cuda"
        self.lv = []
        for lvl in levels:
            ent = {}
            dinv = getattr(lvl, "dinv", None)
            if dinv is not None:
                ent["dinv"] = torch.as_tensor(np.asarray(dinv), device=dev)
                ent["omega"] = float(lvl.omega)
            A_sp = lvl.A.to_scipy_sparse_csr()  # collective gather
            ent["A"] = (ops.LocalCSR.from_scipy(A_sp, dev) if self.on_gpu
                        else A_sp)
            if hasattr(lvl, "Rdown"):
                R_sp = lvl.Rdown.to_scipy_sparse_csr()
                P_sp = lvl.Pdown.to_scipy_sparse_csr()
                if self.on_gpu:
                    ent["R"] = ops.LocalCSR.from_scipy(R_sp, dev)
                    ent["P"] = ops.LocalCSR.from_scipy(P_sp, dev)
                else:
                    ent["R"] = R_sp
                    ent["P"] = P_sp
            self.lv.append(ent)
        self._graph = None
        self._graph_tried = False

    def _spmv(self, M, x):
        if self.on_gpu:
            from .ops import local as ops

            return ops.spmv(M, x)
        return torch.as_tensor(M @ x.numpy())

    def _vcycle(self, i, b):
        lv = self.lv[i]
        if i == len(self.lv) - 1:
            return self.coarse_inv_t @ b
        x = b * lv["dinv"] * lv["omega"]
        for _ in range(self.smooth_iters - 1):
            x = x + lv["omega"] * lv["dinv"] * (b - self._spmv(lv["A"], x))
        r = b - self._spmv(lv["A"], x)
        xc = self._vcycle(i + 1, self._spmv(lv["R"], r))
        x = x + self._spmv(lv["P"], xc)
        for _ in range(self.smooth_iters):
            x = x + lv["omega"] * lv["dinv"] * (b - self._spmv(lv["A"], x))
        return x

    def apply(self, b_full: torch.Tensor) -> torch.Tensor:
        if self.on_gpu and not self._graph_tried:
            self._graph_tried = True
            if not os.environ.get("SPARSE_NO_HIPGRAPH"):
                try:
                    self._gin = b_full.clone()
                    side = torch.cuda.Stream()
                    side.wait_stream(torch.cuda.current_stream())
                    with torch.cuda.stream(side):
                        for _ in range(2):
                            self._vcycle(0, self._gin)
                    torch.cuda.current_stream().wait_stream(side)
                    g = torch.cuda.CUDAGraph()
                    with torch.cuda.graph(g):
                        self._gout = self._vcycle(0, self._gin)
                    self._graph = g
                except Exception as e:
                    print(f"[multigrid] coarse-tail graph capture "
                          f"unavailable ({e}); eager")
                    self._graph = None
        if self._graph is not None:
            self._gin.copy_(b_full)
            self._graph.replay()
            return self._gout
        return self._vcycle(0, b_full)


def find_replication_cut(levels, repl_threshold: int):
    """First level index whose operator is at/below the threshold (len
    if none) — callers replicate levels[cut:]."""
    for i, lvl in enumerate(levels):
        A = lvl.A if hasattr(lvl, "A") else lvl["A"]
        if A.shape[0] <= repl_threshold:
            return i
    return len(levels)
