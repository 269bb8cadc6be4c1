"""Rydberg-atom MIS quantum annealing demo (capability parity with the
reference's quantum benchmark, results/summit/legate_gpu_quantum.out:
driver+cost Hamiltonians over independence sets, Schroedinger evolution via
solve_ivp with the sparse Hamiltonian SpMV).

python examples/quantum_mis.py -l 4 -T 4.0
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import numpy as np

from benchmark import parse_common_args

parser = argparse.ArgumentParser()
parser.add_argument("-l", type=int, default=3, help="lattice edge (l x l grid graph)")
parser.add_argument("-T", type=float, default=4.0, help="anneal time")
parser.add_argument("-rtol", type=float, default=1e-6)
parser.add_argument("-kmax", type=int, default=None,
                    help="truncate the Hilbert space at this excitation "
                         "level (sets of size <= kmax) — the reference "
                         "rydberg benchmark's space-fraction knob; "
                         "REQUIRED for large lattices (l >= 6)")
args, _ = parser.parse_known_args()
_, timer, npx, sparse_mod, linalg, use_sparse = parse_common_args()

import networkx as nx

from sparse import integrate, quantum
from sparse.parallel import comm


def main():
    G = nx.convert_node_labels_to_integers(nx.grid_2d_graph(args.l, args.l))
    ip = quantum.independence_polynomial(G, kmax=args.kmax)
    drv = quantum.HamiltonianDriver(graph=G, energies=(1,), dtype=np.complex128,
                                    kmax=args.kmax)
    mis = quantum.HamiltonianMIS(graph=G, poly=ip, dtype=np.complex128)
    n = mis.nstates
    if comm.rank() == 0:
        print(f"graph {args.l}x{args.l}: {n} independence sets, "
              f"MIS size {mis.mis_size}, ip={ip}")
    Hd = drv.hamiltonian
    Hc = mis.hamiltonian
    T = args.T

    def rhs(t, y):
        s = t / T
        from sparse import asdistarray

        yd = asdistarray(y)
        return (Hd.dot(yd) * (-1j * (1 - s))) + (Hc.dot(yd) * (-1j * s))

    y0 = np.zeros(n, dtype=np.complex128)
    y0[-1] = 1.0  # start in the empty set (driver ground state at s=0)
    timer.start()
    res = integrate.solve_ivp(rhs, (0.0, T), y0, method="DOP853",
                              rtol=args.rtol, atol=1e-9)
    ms = timer.stop()
    state = res.y[:, -1]
    # cost/optimum touch the Hamiltonian's gathered data (collectives):
    # compute on ALL ranks, print on rank 0
    cost = mis.cost_function(state)
    opt = mis.optimum
    ovl = mis.optimum_overlap(state)
    ar = mis.approximation_ratio(state)
    if comm.rank() == 0:
        print(f"evolved in {ms:.1f} ms, {len(res.t) - 1} steps, "
              f"|psi|={np.linalg.norm(state):.6f}")
        print(f"cost <C> = {cost:.4f} "
              f"(optimum {opt:.1f}), "
              f"optimum overlap = {ovl:.4f}, "
              f"approx ratio = {ar:.4f}")


if __name__ == "__main__":
    main()
