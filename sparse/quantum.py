"""Rydberg-atom / maximum-independent-set Hamiltonian tooling.

Reference parity: sparse/quantum.py — HamiltonianDriver builds the sparse
transition Hamiltonian over independence sets incrementally per set size
(quantum.py:27-300), HamiltonianMIS the diagonal cost Hamiltonian
(quantum.py:302-403), enumerate_independent_sets (quantum.py:555-... +
src/quantum/quantum.cc bitset BFS), independence_polynomial
(quantum.py:447-460), raw_create_csr (quantum.py:488-554).

Sets are Python arbitrary-width int bitmasks (the reference uses a C++
IntSet<N> bitset); the enumeration is a host-side BFS over set sizes, and
the Hamiltonian algebra (CSR assembly, SpMV, solve_ivp evolution) runs on
the distributed torch/HIP path.

State ordering matches the reference's MIS convention: states are indexed
with LARGEST sets first and the empty set last (quantum.py:322-325's
flip(repeat(levels, poly))).
"""
from __future__ import annotations

from typing import List

import numpy as np

from .coo import coo_array
from .csr import csr_array

__all__ = [
    "enumerate_independent_sets", "sets_to_sizes", "independence_polynomial",
    "HamiltonianDriver", "HamiltonianMIS", "LegateHamiltonianDriver",
    "LegateHamiltonianMIS", "raw_create_csr",
]


def _neighbor_masks(graph) -> List[int]:
    import networkx as nx  # noqa: F401

    n = graph.number_of_nodes()
    nodes = list(graph.nodes())
    index = {v: i for i, v in enumerate(nodes)}
    nbr = [0] * n
    for u, v in graph.edges():
        iu, iv = index[u], index[v]
        nbr[iu] |= 1 << iv
        nbr[iv] |= 1 << iu
    return nbr


def _popcount64(a: np.ndarray) -> np.ndarray:
    a = a.astype(np.uint64, copy=True)
    out = np.zeros(a.shape, dtype=np.int64)
    while a.any():
        out += (a & np.uint64(1)).astype(np.int64)
        a >>= np.uint64(1)
    return out


def enumerate_independent_sets(graph, k: int, prevk_sets=None, prevk_queues=None):
    """Independence sets of size k from those of size k-1 (BFS expansion,
    reference quantum.cc:27-... with its IntSet bitsets).  Returns
    (sets, queues) bitmask sequences — numpy-vectorized int64 masks for
    graphs with <= 63 nodes, python arbitrary-width ints beyond.

    queues[i] = candidate nodes with index greater than every member of
    sets[i] and not adjacent to it (the canonical-extension frontier)."""
    n = graph.number_of_nodes()
    if n > 63:
        return _enumerate_py(graph, k, prevk_sets, prevk_queues)
    nbr = np.array(_neighbor_masks(graph), dtype=np.int64)
    if k == 1:
        sets = np.array([1 << v for v in range(n)], dtype=np.int64)
        above = np.array([(((1 << n) - 1) >> (v + 1)) << (v + 1) for v in range(n)],
                         dtype=np.int64)
        queues = above & ~nbr
        return sets, queues
    assert prevk_sets is not None and prevk_queues is not None
    S = np.asarray(prevk_sets, dtype=np.int64)
    Q = np.asarray(prevk_queues, dtype=np.int64)
    out_s, out_q = [], []
    # vectorized per candidate node v: every parent whose queue has bit v
    for v in range(n):
        bit = np.int64(1 << v)
        rows = (Q & bit) != 0
        if not rows.any():
            continue
        lowmask = np.int64(((1 << (v + 1)) - 1))
        out_s.append(S[rows] | bit)
        out_q.append(Q[rows] & ~lowmask & ~nbr[v])
    if not out_s:
        return np.zeros(0, dtype=np.int64), np.zeros(0, dtype=np.int64)
    return np.concatenate(out_s), np.concatenate(out_q)


def _enumerate_py(graph, k, prevk_sets, prevk_queues):
    """Arbitrary-width fallback (> 63 nodes)."""
    n = graph.number_of_nodes()
    nbr = _neighbor_masks(graph)
    if k == 1:
        sets = [1 << v for v in range(n)]
        queues = []
        for v in range(n):
            q = 0
            for u in range(v + 1, n):
                if not (nbr[v] >> u) & 1:
                    q |= 1 << u
            queues.append(q)
        return sets, queues
    assert prevk_sets is not None and prevk_queues is not None
    sets, queues = [], []
    for S, Q in zip(prevk_sets, prevk_queues):
        q = Q
        while q:
            v = (q & -q).bit_length() - 1
            q &= q - 1
            sets.append(S | (1 << v))
            queues.append(Q & ~((1 << (v + 1)) - 1) & ~nbr[v])
    return sets, queues


def sets_to_sizes(queues, graph) -> np.ndarray:
    """Popcount of each candidate queue (reference SETS_TO_SIZES)."""
    if isinstance(queues, np.ndarray):
        return _popcount64(queues)
    return np.array([bin(int(q)).count("1") for q in queues], dtype=np.int64)


def independence_polynomial(graph) -> List[int]:
    """ip[k] = number of independence sets of size k (reference
    quantum.py:447-460)."""
    ip = [1]
    sets, nbrs = None, None
    for k in range(1, graph.number_of_nodes() + 1):
        sets, nbrs = enumerate_independent_sets(graph, k, prevk_sets=sets,
                                                prevk_queues=nbrs)
        if len(sets) == 0:
            break
        ip.append(len(sets))
        if isinstance(nbrs, np.ndarray):
            if not nbrs.any():
                break
        elif all(q == 0 for q in nbrs):
            break
    return ip


def raw_create_csr(rows, cols, vals, shape, dtype) -> csr_array:
    """Sorted-coordinate CSR assembly without materializing a COO object
    (reference quantum.py:488-554)."""
    c = coo_array((np.asarray(vals), (np.asarray(rows), np.asarray(cols))),
                  shape=shape, dtype=dtype)
    return c.tocsr()


class HamiltonianDriver:
    """Off-diagonal driver Hamiltonian: couples each independence set S to
    every T = S \\ {v} (one fewer excitation); symmetric, entries
    energies[0] (reference quantum.py:27-300)."""

    def __init__(self, energies: tuple = (1,), graph=None, dtype=np.complex64):
        self.energies = energies
        n = graph.number_of_nodes()
        self.ip = [1]
        all_sets: List[List[int]] = [[0]]
        sets, nbrs = None, None
        for k in range(1, n + 1):
            sets, nbrs = enumerate_independent_sets(graph, k, prevk_sets=sets,
                                                    prevk_queues=nbrs)
            if len(sets) == 0:
                break
            self.ip.append(len(sets))
            all_sets.append(sets)
            if isinstance(nbrs, np.ndarray):
                if not nbrs.any():
                    break
            elif all(q == 0 for q in nbrs):
                break
        self.nstates = sum(self.ip)
        # ascending ids: group k starts at offsets[k]; within each group ids
        # follow VALUE-sorted mask order so subset lookup is a searchsorted
        offsets = np.concatenate([[0], np.cumsum(self.ip)])
        groups = []
        for k, group in enumerate(all_sets):
            g = np.asarray(group, dtype=np.int64) if not isinstance(
                group, np.ndarray) else group
            groups.append(np.sort(g))
        rows_l, cols_l = [], []
        for k in range(1, len(groups)):
            Sk = groups[k]
            sid = offsets[k] + np.arange(len(Sk), dtype=np.int64)
            # peel the k set bits of every mask, vectorized per position
            rem = Sk.copy()
            for _ in range(k):
                low = rem & -rem
                Tm = Sk & ~low  # subset with that member removed
                tid = offsets[k - 1] + np.searchsorted(groups[k - 1], Tm)
                rows_l.append(sid)
                cols_l.append(tid)
                rem = rem & ~low
        rows = np.concatenate(rows_l)
        cols = np.concatenate(cols_l)
        # reference state ordering: largest sets first, empty set last
        rows = self.nstates - 1 - rows
        cols = self.nstates - 1 - cols
        # symmetric: upper + lower halves
        r = np.concatenate([rows, cols])
        c = np.concatenate([cols, rows])
        v = np.full(len(r), energies[0], dtype=dtype)
        self._hamiltonian = raw_create_csr(r, c, v, (self.nstates, self.nstates),
                                           dtype)

    @property
    def hamiltonian(self) -> csr_array:
        return self._hamiltonian


class HamiltonianMIS:
    """Diagonal MIS cost Hamiltonian (reference quantum.py:302-403)."""

    def __init__(self, graph=None, poly=None, energies=(1, 1), dtype=np.complex64):
        if energies == (1, 1):
            energies = (1,)
        self.graph = graph
        self.n = graph.number_of_nodes()
        self.energies = energies
        self.optimization = "max"
        self._is_diagonal = True
        if poly is None:
            poly = independence_polynomial(graph)
        self.nstates = int(np.sum(poly))
        self.dtype = dtype
        self.mis_size = len(poly) - 1
        levels = np.arange(len(poly))
        C = np.flip(np.repeat(levels, poly)).astype(dtype)
        enum_states = np.arange(self.nstates)
        self._hamiltonian = csr_array(
            (C, (enum_states, enum_states)),
            shape=(self.nstates, self.nstates), dtype=dtype)

    @property
    def hamiltonian(self) -> csr_array:
        if self.energies[0] == 1:
            return self._hamiltonian
        return self._hamiltonian * self.energies[0]

    @property
    def _diagonal_hamiltonian(self):
        return self.hamiltonian.data.reshape(-1, 1)

    @property
    def optimum(self):
        return np.max(self._diagonal_hamiltonian.real)

    @property
    def minimum_energy(self):
        return np.min(self._diagonal_hamiltonian.real)

    def cost_function(self, state):
        state = np.asarray(state).reshape(-1, 1)
        return float(np.real(np.conj(state).T @ (self._diagonal_hamiltonian * state)))

    def optimum_overlap(self, state):
        state = np.asarray(state).reshape(-1, 1)
        optimum_indices = np.argwhere(
            self._diagonal_hamiltonian == self.optimum).T[0]
        optimum = np.zeros(self._diagonal_hamiltonian.shape)
        optimum[optimum_indices] = 1
        return float(np.real(np.conj(state).T @ (optimum * state)))

    def approximation_ratio(self, state):
        return self.cost_function(state) / self.optimum


LegateHamiltonianDriver = HamiltonianDriver
LegateHamiltonianMIS = HamiltonianMIS
