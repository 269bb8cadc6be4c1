"""SpMV microbenchmark on an 11-diagonal banded CSR (fp64) — capability
parity with reference examples/dot_microbenchmark.py (347.7
iters/s on 1 V100 at n=10M rows).

python examples/dot_microbenchmark.py -n 10000000 -iters 300
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import numpy as np

from benchmark import parse_common_args

parser = argparse.ArgumentParser()
parser.add_argument("-n", type=int, default=2**24)
parser.add_argument("-iters", "-i", type=int, default=300, dest="iters")
parser.add_argument("-warmup", type=int, default=10)
parser.add_argument("-nnz-per-row", type=int, default=11, dest="nnz_per_row")
parser.add_argument("-op", choices=["spmv", "spmm"], default="spmv")
parser.add_argument("-k", type=int, default=32, help="B columns for -op spmm")
args, _ = parser.parse_known_args()
_, timer, npx, sparse, linalg, use_sparse = parse_common_args()

if use_sparse:
    from sparse import darray, gallery
    from sparse.parallel import comm

    A = gallery.banded(args.n, ndiags=args.nnz_per_row)
    if args.op == "spmm":
        x = darray.ones((args.n, args.k), dtype=np.float64)
        y = darray.zeros((args.n, args.k), dtype=np.float64)
    else:
        x = darray.ones((args.n,), dtype=np.float64)
        y = darray.zeros((args.n,), dtype=np.float64)
    for _ in range(args.warmup):
        A.dot(x, out=y)
    timer.start()
    for _ in range(args.iters):
        A.dot(x, out=y)
    ms = timer.stop()
    nnz = A.nnz  # collective: all ranks
    if comm.rank() == 0:
        kk = args.k if args.op == "spmm" else 1
        gflops = 2.0 * nnz * kk * args.iters / (ms / 1000.0) / 1e9
        print(f"{args.iters} {args.op}s in {ms:.1f} ms "
              f"({args.iters / (ms / 1000.0):.2f} iters/s, {gflops:.1f} GFLOP/s)")
else:
    import scipy.sparse as sps

    hw = args.nnz_per_row // 2
    offs = list(range(-hw, args.nnz_per_row - hw))
    A = sps.diags([np.ones(args.n)] * args.nnz_per_row, offs,
                  (args.n, args.n)).tocsr()
    x = np.ones(args.n)
    timer.start()
    for _ in range(args.iters):
        y = A @ x
    ms = timer.stop()
    print(f"{args.iters} SpMVs in {ms:.1f} ms ({args.iters / (ms / 1000.0):.2f} iters/s)")
