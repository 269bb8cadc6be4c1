"""End-to-end example runs on the GPU (small sizes; @gpu)."""
import os
import subprocess
import sys

import pytest

pytestmark = pytest.mark.gpu
EX = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "examples")


def run(script, *a, timeout=300):
    r = subprocess.run([sys.executable, os.path.join(EX, script), *a],
                       capture_output=True, timeout=timeout)
    assert r.returncode == 0, r.stdout.decode()[-1200:] + r.stderr.decode()[-1200:]
    return r.stdout.decode()


def test_pde_gpu():
    out = run("pde.py", "-nx", "512", "-ny", "512", "-throughput", "-max_iter", "100")
    assert "Solve finished: 100 iterations" in out


def test_gmg_gpu():
    out = run("gmg.py", "-N", "255", "-maxiter", "50")
    assert "info=0" in out


def test_amg_gpu():
    out = run("amg.py", "-n", "65536", "-maxiter", "100")
    assert "info=0" in out


def test_dot_micro_gpu():
    out = run("dot_microbenchmark.py", "-n", "1000000", "-iters", "20", "-warmup", "3")
    assert "spmvs" in out


def test_gmg_variants_gpu():
    out = run("gmg.py", "-N", "127", "-data", "diffusion")
    assert "info=0" in out
    out = run("gmg.py", "-N", "127", "-smoother", "symgs")
    assert "info=0" in out
    out = run("gmg.py", "-N", "127", "-gridop", "injection")
    assert "info=0" in out


def test_spmm_micro_bsr_gpu():
    """dot_microbenchmark -op spmm engages the BSR-MFMA route on the
    banded matrix (and the off-switch produces the same numbers)."""
    out = run("dot_microbenchmark.py", "-op", "spmm", "-k", "32",
              "-n", "500000", "-iters", "10", "-warmup", "2")
    assert "spmms" in out


def test_quantum_kmax_gpu():
    out = run("quantum_mis.py", "-l", "6", "-T", "1.0", "-kmax", "6",
              timeout=400)
    assert "approx ratio" in out


def test_gmg_3d_gpu():
    out = run("gmg.py", "-N", "127", "-dim", "3", timeout=400)
    assert "info=0" in out
