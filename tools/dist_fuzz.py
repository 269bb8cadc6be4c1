import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, scipy.sparse as sps, torch
import torch.distributed as dist

def main():
    dist.init_process_group("gloo")
    import sparse
    from sparse import csr_array, csc_array, linalg
    rng = np.random.default_rng(314159)  # same on all ranks -> same trials
    fails = 0
    for trial in range(150):
        m, n, k = [int(v) for v in rng.integers(1, 45, 3)]
        dt = [np.float64, np.complex128][trial % 2]
        a = sps.random(m, n, rng.random()*0.5, random_state=int(rng.integers(1e6)), format="csr").astype(dt)
        if dt == np.complex128:
            a.data = a.data + 1j * rng.random(a.nnz)
        a.sort_indices()
        A = csr_array(a)
        x = rng.random(n)
        try:
            assert np.allclose(np.asarray(A @ x), a @ x, atol=1e-8), "spmv"
            b2 = sps.random(n, k, 0.4, random_state=int(rng.integers(1e6)), format="csr").astype(dt)
            C = A @ csr_array(b2)
            assert np.allclose(np.asarray(C.todense()), (a @ b2).toarray(), atol=1e-8), "spgemm"
            assert np.allclose(np.asarray(A.tocsc().todense()), a.toarray(), atol=1e-10), "tocsc"
            assert np.allclose(np.asarray(A.T.tocsr().todense()), a.T.toarray(), atol=1e-10), "T"
            co = A.tocoo()
            assert np.allclose(np.asarray(co.tocsr().todense()), a.toarray(), atol=1e-10), "coo rt"
            a2 = sps.random(m, n, 0.3, random_state=int(rng.integers(1e6)), format="csr").astype(dt)
            assert np.allclose(np.asarray((A + csr_array(a2)).todense()), (a + a2).toarray(), atol=1e-8), "add"
            A.balance()
            assert np.allclose(np.asarray(A @ x), a @ x, atol=1e-8), "balanced spmv"
            if trial % 7 == 0:
                d = A[min(2, m-1):m]
                assert np.allclose(np.asarray(d.todense()), a[min(2, m-1):m].toarray(), atol=1e-10), "slice"
        except AssertionError as e:
            fails += 1
            print(f"rank{dist.get_rank()} FAIL trial {trial}: {e} {(m,n,k)} {dt}", flush=True)
            if fails > 3: break
    t = torch.tensor([fails]); dist.all_reduce(t)
    if dist.get_rank() == 0:
        print("DIST_FUZZ", "ALL OK (150 trials)" if t.item() == 0 else f"{t.item()} FAILURES", flush=True)
    dist.destroy_process_group()

main()
