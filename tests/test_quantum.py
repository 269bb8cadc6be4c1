"""Quantum MIS module tests vs networkx-derived oracles."""
import numpy as np
import pytest

nx = pytest.importorskip("networkx")

from sparse import quantum


def brute_force_independent_sets(G):
    nodes = list(G.nodes())
    n = len(nodes)
    adj = {i: set() for i in range(n)}
    idx = {v: i for i, v in enumerate(nodes)}
    for u, v in G.edges():
        adj[idx[u]].add(idx[v])
        adj[idx[v]].add(idx[u])
    sets_by_size = {}
    for mask in range(1 << n):
        members = [i for i in range(n) if mask >> i & 1]
        ok = all(j not in adj[i] for i in members for j in members)
        if ok:
            sets_by_size.setdefault(len(members), []).append(mask)
    return sets_by_size


@pytest.mark.parametrize("G", [
    nx.path_graph(5), nx.cycle_graph(6), nx.complete_graph(4),
    nx.grid_2d_graph(3, 3),
])
def test_independence_polynomial(G):
    G = nx.convert_node_labels_to_integers(G)
    ip = quantum.independence_polynomial(G)
    bf = brute_force_independent_sets(G)
    expect = [len(bf.get(k, [])) for k in range(max(bf) + 1)]
    assert ip == expect, (ip, expect)


def test_enumerate_matches_bruteforce():
    G = nx.convert_node_labels_to_integers(nx.cycle_graph(7))
    bf = brute_force_independent_sets(G)
    sets, queues = quantum.enumerate_independent_sets(G, 1)
    assert sorted(sets) == sorted(bf[1])
    sets2, _ = quantum.enumerate_independent_sets(G, 2, sets, queues)
    assert sorted(sets2) == sorted(bf[2])


def test_hamiltonian_driver_structure():
    G = nx.convert_node_labels_to_integers(nx.path_graph(4))
    drv = quantum.HamiltonianDriver(graph=G, energies=(1,))
    H = np.asarray(drv.hamiltonian.todense())
    # symmetric with zero diagonal
    assert np.allclose(H, H.T)
    assert np.allclose(np.diag(H), 0)
    # every size-k set connects to exactly k subsets of size k-1
    ip = quantum.independence_polynomial(G)
    nstates = sum(ip)
    assert H.shape == (nstates, nstates)
    # total edges (one direction) = sum over sets of their size
    sets_total = sum(k * c for k, c in enumerate(ip))
    assert (H != 0).sum() == 2 * sets_total


def test_hamiltonian_mis_diagonal():
    G = nx.convert_node_labels_to_integers(nx.cycle_graph(5))
    ip = quantum.independence_polynomial(G)
    mis = quantum.HamiltonianMIS(graph=G, poly=ip)
    d = mis._diagonal_hamiltonian.ravel()
    # largest sets first, empty set last
    assert d[0] == mis.mis_size
    assert d[-1] == 0
    assert mis.optimum == mis.mis_size
    # uniform superposition cost = average set size
    state = np.ones(mis.nstates) / np.sqrt(mis.nstates)
    expect = sum(k * c for k, c in enumerate(ip)) / mis.nstates
    assert np.isclose(mis.cost_function(state), expect)


def test_annealing_evolution_small():
    """Schroedinger evolution i dy/dt = -H(t) y with solve_ivp (the
    quantum app's driver loop shape)."""
    from sparse import integrate

    G = nx.convert_node_labels_to_integers(nx.path_graph(3))
    ip = quantum.independence_polynomial(G)
    drv = quantum.HamiltonianDriver(graph=G, energies=(1,), dtype=np.complex128)
    mis = quantum.HamiltonianMIS(graph=G, poly=ip, dtype=np.complex128)
    Hd = np.asarray(drv.hamiltonian.todense())
    Hc = np.asarray(mis.hamiltonian.todense())
    T = 2.0

    def rhs(t, y):
        s = t / T
        H = (1 - s) * Hd + s * Hc
        return -1j * (H @ np.asarray(y))

    y0 = np.zeros(mis.nstates, dtype=np.complex128)
    y0[-1] = 1.0  # empty set
    res = integrate.solve_ivp(rhs, (0, T), y0, method="DOP853", rtol=1e-8,
                              atol=1e-10)
    assert res.success
    # norm conserved
    assert np.isclose(np.linalg.norm(res.y[:, -1]), 1.0, atol=1e-6)


def test_limbs_path_matches_int64():
    """Force the multi-limb (>63-node) enumeration on a small graph and
    compare sets/queues/polynomial/driver against the int64 path."""
    import networkx as nx

    from sparse import quantum

    g = nx.random_geometric_graph(18, 0.45, seed=5)
    ip_fast = quantum.independence_polynomial(g)
    H_fast = quantum.HamiltonianDriver(graph=g).hamiltonian.to_scipy_sparse_csr()
    old = quantum._INT64_MAX_NODES
    quantum._INT64_MAX_NODES = 0
    try:
        ip_limbs = quantum.independence_polynomial(g)
        H_limbs = quantum.HamiltonianDriver(graph=g).hamiltonian.to_scipy_sparse_csr()
    finally:
        quantum._INT64_MAX_NODES = old
    assert ip_fast == ip_limbs
    assert (H_fast != H_limbs).nnz == 0


def test_wide_graph_polynomial():
    """>63-node graph (multi-limb path): disjoint union of 5 cliques of 14
    nodes (70 total) — an independent set picks at most one node per
    clique, so ip[k] = C(5,k) * 14^k."""
    import math

    import networkx as nx

    from sparse import quantum

    g = nx.disjoint_union_all([nx.complete_graph(14) for _ in range(5)])
    assert g.number_of_nodes() == 70
    ip = quantum.independence_polynomial(g)
    assert len(ip) == 6
    for k in range(6):
        assert ip[k] == math.comb(5, k) * 14 ** k, k


def test_enum_budget_guard_and_kmax():
    """Unbounded enumeration of a large lattice must raise (a 9x9 grid has
    ~1e14 independence sets — it took a GPU HOST down before this guard);
    kmax truncation keeps it tractable and consistent."""
    import networkx as nx

    from sparse import quantum

    G = nx.convert_node_labels_to_integers(nx.grid_2d_graph(7, 7))
    with pytest.raises(ValueError, match="exceeded"):
        quantum.independence_polynomial(G, max_states=50_000)
    ip = quantum.independence_polynomial(G, kmax=3)
    assert len(ip) == 4 and ip[0] == 1 and ip[1] == 49
    drv = quantum.HamiltonianDriver(graph=G, kmax=3, dtype=np.complex128)
    assert drv.nstates == sum(ip)
    H = drv.hamiltonian
    assert H.shape == (drv.nstates, drv.nstates)
    # driver couples size-k sets to size-(k-1) subsets: nnz = 2 * sum k*ip[k]
    assert H.nnz == 2 * sum(k * c for k, c in enumerate(ip))
