"""Run the examples end-to-end as tests (reference test.py runs examples
too, test.py:27-30)."""
import os
import subprocess
import sys

import pytest

EX = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "examples")


def run(script, *a, timeout=420):
    r = subprocess.run([sys.executable, os.path.join(EX, script), *a],
                       capture_output=True, timeout=timeout)
    assert r.returncode == 0, r.stdout.decode()[-1500:] + r.stderr.decode()[-1500:]
    return r.stdout.decode()


def test_pde():
    out = run("pde.py", "-nx", "64", "-ny", "64", "-tol", "1e-8")
    assert "Solve finished" in out


def test_pde_throughput():
    out = run("pde.py", "-nx", "64", "-ny", "64", "-throughput", "-max_iter", "50")
    assert "Solve finished: 50 iterations" in out


def test_dot_microbenchmark():
    out = run("dot_microbenchmark.py", "-n", "100000", "-iters", "5", "-warmup", "1")
    assert "SpMVs" in out


def test_gmg():
    out = run("gmg.py", "-N", "127", "-maxiter", "50")
    assert "info=0" in out
    iters = int(out.split("iters=")[1].split()[0])
    assert iters <= 12, out  # multigrid must converge fast


def test_amg():
    out = run("amg.py", "-n", "16384", "-maxiter", "200")
    assert "info=0" in out
    iters = int(out.split("iters=")[1].split()[0])
    assert iters <= 40, out


def test_spgemm_microbenchmark():
    out = run("spgemm_microbenchmark.py", "-nx", "129", "-iters", "2", "-warmup", "1")
    assert "R@A@P" in out


def test_spectral_norm():
    out = run("spectral_norm.py")
    assert "OK" in out


def test_gmg_3d():
    out = run("gmg.py", "-N", "31", "-dim", "3", "-maxiter", "60")
    assert "info=0" in out
    iters = int(out.split("iters=")[1].split()[0])
    assert iters <= 15, out


def test_quantum_mis():
    out = run("quantum_mis.py", "-l", "3", "-T", "2.0")
    assert "approx ratio" in out
    ratio = float(out.split("approx ratio = ")[1].split()[0])
    assert 0.3 < ratio <= 1.0
