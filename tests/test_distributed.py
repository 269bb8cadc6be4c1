"""Multi-process distributed tests (gloo backend, CPU — runs in CI).

The same code paths carry the 8-GPU RCCL runs; SURVEY §4's "distributed
testing via more processors" obligation.
"""
import os
import subprocess
import sys

import pytest


@pytest.mark.parametrize("nproc", [2, 3, 4])
def test_distributed_battery(nproc):
    """ws=3 covers odd world sizes: uneven slabs, 3x1 2-D SpGEMM grids,
    asymmetric alltoallv counts."""
    env = dict(os.environ)
    env.update({
        "MASTER_ADDR": "127.0.0.1",
        "MASTER_PORT": str(29650 + nproc),
        "GLOO_SOCKET_IFNAME": env.get("GLOO_SOCKET_IFNAME", "lo"),
    })
    worker = os.path.join(os.path.dirname(__file__), "dist_worker.py")
    procs = []
    for r in range(nproc):
        e = dict(env)
        e.update({"RANK": str(r), "WORLD_SIZE": str(nproc), "LOCAL_RANK": str(r)})
        procs.append(subprocess.Popen(
            [sys.executable, worker], env=e,
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
            cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__)))))
    outs = []
    ok = True
    for p in procs:
        try:
            out, _ = p.communicate(timeout=300)
        except subprocess.TimeoutExpired:
            p.kill()
            out, _ = p.communicate()
            ok = False
        outs.append(out.decode(errors="replace"))
        ok = ok and p.returncode == 0
    assert ok, "\n".join(outs[-2:])
    assert any("DIST_ALL_OK" in o for o in outs), outs[0][-2000:]


@pytest.mark.parametrize("nproc,extra", [(2, []), (2, ["--weak"]), (8, [])])
def test_bench_multirank(nproc, extra):
    """bench.py --gpus N must be a tested code path:
    run it at ws=2 and ws=8 (the driver's SCALE shape) on gloo/CPU and
    require the one-line JSON contract."""
    import json

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ)
    env.update({"MASTER_ADDR": "127.0.0.1",
                "MASTER_PORT": str(29810 + nproc + len(extra)),
                "GLOO_SOCKET_IFNAME": env.get("GLOO_SOCKET_IFNAME", "lo"),
                "WORLD_SIZE": str(nproc)})
    nx = "128" if nproc == 2 else "64"
    argv = [sys.executable, os.path.join(root, "bench.py"),
            "--gpus", str(nproc), "--steps", "4", "--warmup", "1",
            "--nx", nx, *extra]
    procs = []
    for r in range(nproc):
        e = dict(env)
        e.update({"RANK": str(r), "LOCAL_RANK": str(r)})
        procs.append(subprocess.Popen(argv, env=e, stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT, cwd=root))
    outs = []
    for p in procs:
        out, _ = p.communicate(timeout=300)
        outs.append(out.decode(errors="replace"))
        assert p.returncode == 0, outs[-1][-1500:]
    line = [ln for ln in outs[0].splitlines() if ln.startswith("{")]
    assert len(line) == 1, outs[0][-800:]
    rec = json.loads(line[0])
    assert rec["metric"] == "cg_iters_per_sec" and rec["n_gpus"] == nproc
    assert rec["scaling"] == ("weak" if extra else "strong")
    if extra:  # sqrt(ws)-scaled grid edge
        assert rec["config"]["nx"] == int(round(int(nx) * nproc ** 0.5))
    # other ranks stay silent (one JSON line per job)
    assert not any(ln.startswith("{") for o in outs[1:] for ln in o.splitlines())


def test_examples_multirank():
    """The examples themselves must be SPMD-clean (no rank-0-only
    collectives): run the main ones at ws=2 on gloo."""
    ex = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                      "examples")
    cases = [
        ("pde.py", ["-nx", "120", "-ny", "120", "-throughput", "-max_iter", "30"],
         "residual norm"),
        ("gmg.py", ["-N", "31"], "info=0"),
        # replicated-coarse-tail boundary: fine level distributed, levels
        # >= 1 replicated (VERDICT r1 #3 multi-GPU V-cycle latency plan)
        ("gmg.py", ["-N", "127", "-repl_threshold", "5000"], "info=0"),
        ("amg.py", ["-n", "1024"], "info=0"),
        ("dot_microbenchmark.py", ["-n", "50000", "-iters", "3", "-warmup", "1"],
         "spmvs"),
        ("spgemm_microbenchmark.py", ["-nx", "63", "-iters", "2"], "R@A@P"),
        ("quantum_mis.py", ["-l", "3", "-T", "1.0"], "approx ratio"),
    ]
    for i, (script, argv, needle) in enumerate(cases):
        env = dict(os.environ)
        env.update({"MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(29930 + i),
                    "GLOO_SOCKET_IFNAME": env.get("GLOO_SOCKET_IFNAME", "lo"),
                    "WORLD_SIZE": "2"})
        procs = []
        for r in range(2):
            e = dict(env)
            e.update({"RANK": str(r), "LOCAL_RANK": str(r)})
            procs.append(subprocess.Popen(
                [sys.executable, os.path.join(ex, script), *argv], env=e,
                stdout=subprocess.PIPE, stderr=subprocess.STDOUT))
        outs = []
        for p in procs:
            out, _ = p.communicate(timeout=300)
            outs.append(out.decode(errors="replace"))
            assert p.returncode == 0, f"{script}: {outs[-1][-1200:]}"
        assert needle in outs[0], f"{script}: {outs[0][-400:]}"
