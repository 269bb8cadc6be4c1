"""Track down the large-n nan in the fused CG loop: standalone kernel checks
at n=36M plus an rz trace of the manual fused iteration."""
import math
import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch

from sparse import darray, gallery, linalg
from sparse import kernels

kernels.require()
n = 36_000_000

# 1. axpby_norm2 standalone at 36M
y = darray.random((n,), seed=1)
x = darray.random((n,), seed=2)
y0 = y.local.clone()
a = torch.tensor(3.14, device="cuda", dtype=torch.float64)
b = torch.tensor(2.7, device="cuda", dtype=torch.float64)
rz = kernels.axpby_norm2(y.local, x.local, a, b, True, True)
ref = y0 - (3.14 / 2.7) * x.local
print("axpby_norm2: ydiff", float((y.local - ref).abs().max()),
      "rz", float(rz.item()), "ref", float((ref * ref).sum().item()), flush=True)

# 2. cg_axpby standalone at 36M (both isalpha modes)
y = darray.DistArray.from_local(y0.clone(), x.partition, (n,))
kernels.axpby(y.local, x.local, a, b, True, False)
print("axpby alpha: diff", float((y.local - (y0 + (3.14/2.7) * x.local)).abs().max()), flush=True)
y2 = y0.clone()
kernels.axpby(y2, x.local, a, b, False, False)
print("axpby beta: diff", float((y2 - ((3.14/2.7) * y0 + x.local)).abs().max()), flush=True)

# 3. manual fused loop at nx=6000, trace rz/pq
nx = 6000
hx = 1.0 / (nx - 1)
A = gallery.poisson2d(nx, nx, scale=1.0 / (hx * hx))
m = A.shape[0]
ii = darray.arange(m).astype(np.float64)
xc = (ii.local % nx).to(torch.float64) * hx
yc = (ii.local // nx).to(torch.float64) * hx
bl = torch.sin(math.pi * xc) * torch.sin(math.pi * yc)
bv = darray.DistArray.from_local(bl, ii.partition, (m,))

from sparse.linalg import cg_axpby, _axpby_norm2
xs = darray.zeros((m,), dtype=np.float64)
r = bv - A.dot(xs)
p = r.copy()
rz = r.dot(r)
q = darray.zeros((m,), dtype=np.float64)
for i in range(300):
    pq = A.spmv_dot(p, q)
    cg_axpby(xs, p, rz, pq, isalpha=True, negate=False)
    rz_new = _axpby_norm2(r, q, rz, pq, negate=True)
    if i % 20 == 0 or i < 5:
        rzv = float(rz.item()); pqv = float(pq.item()); rznv = float(rz_new.item())
        print(f"i={i} rz={rzv:.6e} pq={pqv:.6e} rz_new={rznv:.6e}", flush=True)
        if math.isnan(rznv):
            print("r nan count:", int(torch.isnan(r.local).sum()),
                  "q nan:", int(torch.isnan(q.local).sum()),
                  "p nan:", int(torch.isnan(p.local).sum()),
                  "x nan:", int(torch.isnan(xs.local).sum()), flush=True)
            break
    cg_axpby(p, r, rz_new, rz, isalpha=False, negate=False)
    rz = rz_new
print("final rz:", float(rz.item()), flush=True)
