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

from typing import List, Optional

import numpy as np

from .coo import coo_array
from .csr import csr_array

__all__ = [
    "enumerate_independent_sets", "sets_to_sizes", "independence_polynomial",
    "HamiltonianDriver", "HamiltonianMIS", "LegateHamiltonianDriver",
    "LegateHamiltonianMIS", "raw_create_csr",
]


# graphs wider than this use the multi-limb (n, L) uint64 path; tests
# lower it to force limb coverage on small graphs
_INT64_MAX_NODES = 63


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


def _neighbor_limbs(graph, L: int) -> np.ndarray:
    """(n, L) uint64 adjacency bitsets, limb 0 = bits 0..63."""
    n = graph.number_of_nodes()
    nodes = list(graph.nodes())
    index = {v: i for i, v in enumerate(nodes)}
    nbr = np.zeros((n, L), dtype=np.uint64)
    for u, v in graph.edges():
        iu, iv = index[u], index[v]
        nbr[iu, iv // 64] |= np.uint64(1) << np.uint64(iv % 64)
        nbr[iv, iu // 64] |= np.uint64(1) << np.uint64(iu % 64)
    return nbr


def _limb_sortable(masks: np.ndarray) -> np.ndarray:
    """View (m, L) uint64 limb masks as a structured array whose record
    comparison is lexicographic most-significant-limb first — sortable and
    searchsorted-compatible."""
    L = masks.shape[1]
    rev = np.ascontiguousarray(masks[:, ::-1])
    dt = np.dtype([("f%d" % i, "<u8") for i in range(L)])
    return rev.view(dt).reshape(masks.shape[0])


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
    if n > _INT64_MAX_NODES:
        return _enumerate_limbs(graph, k, prevk_sets, prevk_queues)
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


def _enumerate_limbs(graph, k, prevk_sets, prevk_queues):
    """Vectorized arbitrary-width path (> 63 nodes): masks are (m, L)
    uint64 limb arrays, L = ceil(n/64) — the MI355X counterpart of the
    reference's templated IntSet<N,T> bitsets (quantum.h:27-...)."""
    n = graph.number_of_nodes()
    L = (n + 63) // 64
    nbr = _neighbor_limbs(graph, L)
    one = np.uint64(1)
    if k == 1:
        sets = np.zeros((n, L), dtype=np.uint64)
        above = np.zeros((n, L), dtype=np.uint64)
        for v in range(n):
            li, sh = divmod(v, 64)
            sets[v, li] = one << np.uint64(sh)
            above[v, li] = (~np.uint64(0)) << np.uint64(sh)
            above[v, li] &= ~(one << np.uint64(sh))
            above[v, li + 1:] = ~np.uint64(0)
            if n % 64:
                above[v, L - 1] &= (one << np.uint64(n % 64)) - one
        queues = above & ~nbr
        return sets, queues
    assert prevk_sets is not None and prevk_queues is not None
    S = np.asarray(prevk_sets, dtype=np.uint64)
    Q = np.asarray(prevk_queues, dtype=np.uint64)
    out_s, out_q = [], []
    for v in range(n):
        li, sh = divmod(v, 64)
        rows = (Q[:, li] >> np.uint64(sh)) & one != 0
        if not rows.any():
            continue
        ns = S[rows].copy()
        ns[:, li] |= one << np.uint64(sh)
        lowmask = np.zeros(L, dtype=np.uint64)
        lowmask[:li] = ~np.uint64(0)
        lowmask[li] = ((one << np.uint64(sh)) - one) | (one << np.uint64(sh))
        nq = Q[rows] & ~lowmask & ~nbr[v]
        out_s.append(ns)
        out_q.append(nq)
    if not out_s:
        z = np.zeros((0, L), dtype=np.uint64)
        return z, z.copy()
    return np.concatenate(out_s), np.concatenate(out_q)


def sets_to_sizes(queues, graph) -> np.ndarray:
    """Popcount of each candidate queue (reference SETS_TO_SIZES)."""
    if isinstance(queues, np.ndarray):
        if queues.ndim == 2:  # multi-limb
            return _popcount64(queues).sum(axis=1)
        return _popcount64(queues)
    return np.array([bin(int(q)).count("1") for q in queues], dtype=np.int64)


# hard ceiling on enumerated states: grid-graph counts grow like 1.5^n —
# an unbounded l=9 (81-node) enumeration is ~1e14 sets and takes the HOST
# down with it.  Raise a clear error instead (override via kmax or
# max_states).
MAX_ENUM_STATES = 1 << 27


def _check_enum_budget(total: int, k: int, max_states: int) -> None:
    if total > max_states:
        raise ValueError(
            f"independence-set enumeration exceeded {max_states} states at "
            f"size k={k} ({total} so far); this graph's Hilbert space is "
            f"intractable to enumerate fully. Truncate with kmax= (the "
            f"excitation-level cutoff; quantum_mis.py -kmax) or raise "
            f"max_states explicitly.")


def independence_polynomial(graph, kmax: Optional[int] = None,
                            max_states: int = MAX_ENUM_STATES) -> List[int]:
    """ip[k] = number of independence sets of size k (reference
    quantum.py:447-460).  kmax truncates enumeration at that set size
    (the Hilbert-space-fraction knob of the reference's external rydberg
    benchmark)."""
    ip = [1]
    sets, nbrs = None, None
    hi = graph.number_of_nodes() if kmax is None else min(
        kmax, graph.number_of_nodes())
    for k in range(1, hi + 1):
        sets, nbrs = enumerate_independent_sets(graph, k, prevk_sets=sets,
                                                prevk_queues=nbrs)
        if len(sets) == 0:
            break
        ip.append(len(sets))
        _check_enum_budget(sum(ip), k, max_states)
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

    def __init__(self, energies: tuple = (1,), graph=None, dtype=np.complex64,
                 kmax: Optional[int] = None,
                 max_states: int = MAX_ENUM_STATES):
        self.energies = energies
        n = graph.number_of_nodes()
        self.ip = [1]
        all_sets: List[List[int]] = [[0]]
        sets, nbrs = None, None
        hi = n if kmax is None else min(kmax, n)
        for k in range(1, hi + 1):
            sets, nbrs = enumerate_independent_sets(graph, k, prevk_sets=sets,
                                                    prevk_queues=nbrs)
            if len(sets) == 0:
                break
            self.ip.append(len(sets))
            all_sets.append(sets)
            _check_enum_budget(sum(self.ip), k, max_states)
            if isinstance(nbrs, np.ndarray):
                if not nbrs.any():
                    break
            elif all(q == 0 for q in nbrs):
                break
        self.nstates = sum(self.ip)
        # ascending ids: group k starts at offsets[k]; within each group ids
        # follow VALUE-sorted mask order so subset lookup is a searchsorted
        offsets = np.concatenate([[0], np.cumsum(self.ip)])
        limbs = any(isinstance(g, np.ndarray) and g.ndim == 2
                    for g in all_sets[1:])
        if limbs:
            L = next(g.shape[1] for g in all_sets[1:]
                     if isinstance(g, np.ndarray) and g.ndim == 2)
            all_sets[0] = np.zeros((1, L), dtype=np.uint64)
        groups, sortables = [], []
        for k, group in enumerate(all_sets):
            if limbs:
                g = np.asarray(group, dtype=np.uint64)
                key = _limb_sortable(g)
                order = np.argsort(key)
                groups.append(g[order])
                sortables.append(key[order])
            else:
                g = np.asarray(group, dtype=np.int64) if not isinstance(
                    group, np.ndarray) else group
                groups.append(np.sort(g))
                sortables.append(groups[-1])
        rows_l, cols_l = [], []
        for k in range(1, len(groups)):
            Sk = groups[k]
            sid = offsets[k] + np.arange(len(Sk), dtype=np.int64)
            # peel the k set bits of every mask, vectorized per position
            rem = Sk.copy()
            for _ in range(k):
                if limbs:
                    m = len(Sk)
                    ar = np.arange(m)
                    li = np.argmax(rem != 0, axis=1)
                    limbv = rem[ar, li]
                    low = limbv & (np.uint64(0) - limbv)
                    Tm = Sk.copy()
                    Tm[ar, li] = Sk[ar, li] & ~low
                    tid = offsets[k - 1] + np.searchsorted(
                        sortables[k - 1], _limb_sortable(Tm))
                    rem[ar, li] = limbv & ~low
                else:
                    low = rem & -rem
                    Tm = Sk & ~low  # subset with that member removed
                    tid = offsets[k - 1] + np.searchsorted(groups[k - 1], Tm)
                    rem = rem & ~low
                rows_l.append(sid)
                cols_l.append(tid)
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
