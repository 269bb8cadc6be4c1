"""Bisect the pde nx=6000 throughput-mode nan: DIA vs ELL vs CSR paths."""
import math
import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch

from sparse import gallery, darray, linalg

nx = 6000
hx = 1.0 / (nx - 1)

def mk():
    return gallery.poisson2d(nx, nx, scale=1.0 / (hx * hx))

A = mk()
n = A.shape[0]
ii = darray.arange(n).astype(np.float64)
x_coord = (ii.local % nx).to(A._values.dtype) * hx
y_coord = (ii.local // nx).to(A._values.dtype) * hx
blocal = torch.sin(math.pi * x_coord) * torch.sin(math.pi * y_coord)
b = darray.DistArray.from_local(blocal, ii.partition, (n,))

x = darray.random((n,), seed=7)
y1 = A @ x
print("dia used:", A._dia() is not None, flush=True)
A2 = mk(); A2._dia_cache = "no"; A2._ell_cache = "no"
y2 = A2 @ x
print("spmv dia-vs-csr inf diff:", float((y1.local - y2.local).abs().max()), flush=True)
q = darray.zeros((n,))
dot = A.spmv_dot(x, q)
print("fused dot:", float(dot), "ref:", float(torch.dot(x.local, y2.local)),
      "q diff:", float((q.local - y2.local).abs().max()), flush=True)

for tag, Ax in [("dia", A), ("ell", None), ("csr", A2)]:
    if tag == "ell":
        Ax = mk(); Ax._dia_cache = "no"
        print("ell used:", Ax._ell() is not None, flush=True)
    xs, info = linalg.cg(Ax, b, tol=0.0, atol=0.0, maxiter=300,
                         conv_test_iters=None)
    r = b - Ax.dot(xs)
    print(f"cg300 {tag}: resid={float(r.norm().item()):.6e} "
          f"nan_in_x={bool(torch.isnan(xs.local).any())}", flush=True)
