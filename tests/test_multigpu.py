"""Multi-GPU RCCL battery — picked up by the driver's GPU test tier on a
multi-GPU box; auto-skipped on 1-GPU boxes.

Runs the SAME SPMD battery as the gloo CI tests, but one process per GPU
over RCCL/xGMI: this exercises the nccl branches of comm.all_to_all_v
(batched isend/irecv op ordering), gather_halos_begin/end (overlap), the
complex view-as-real wire format, and the weak-scaled bench path on real
hardware.  Reference protocol: results/summit/legate_gpu_pde.out.
"""
import json
import os
import subprocess
import sys

import pytest
import torch

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _ngpus() -> int:
    return torch.cuda.device_count() if torch.cuda.is_available() else 0


def _launch_spmd(argv, nproc, port, timeout=600):
    env = dict(os.environ)
    env.update({
        "MASTER_ADDR": "127.0.0.1",
        "MASTER_PORT": str(port),
        "WORLD_SIZE": str(nproc),
        "SPARSE_DIST_BACKEND": "nccl",
    })
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    procs = []
    for r in range(nproc):
        e = dict(env)
        e.update({"RANK": str(r), "LOCAL_RANK": str(r)})
        procs.append(subprocess.Popen(argv, env=e, stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT, cwd=ROOT))
    outs, rcs = [], []
    for p in procs:
        try:
            out, _ = p.communicate(timeout=timeout)
        except subprocess.TimeoutExpired:
            p.kill()
            out, _ = p.communicate()
        outs.append(out.decode(errors="replace"))
        rcs.append(p.returncode)
    return rcs, outs


@pytest.mark.gpu
def test_gloo_on_gpu_battery():
    """ws=2 battery with BOTH ranks sharing one GPU over gloo (host-staged
    wire): runs every GPU kernel under world_size>1 on a 1-GPU box — the
    halo interior/boundary split, plan edges and fused CG paths execute
    with device tensors even when no multi-GPU node is available."""
    if _ngpus() < 1:
        pytest.skip("needs a GPU")
    worker = os.path.join(ROOT, "tests", "dist_worker.py")
    env = dict(os.environ)
    env.update({"MASTER_ADDR": "127.0.0.1", "MASTER_PORT": "29745",
                "WORLD_SIZE": "2", "SPARSE_DIST_BACKEND": "gloo",
                "GLOO_SOCKET_IFNAME": env.get("GLOO_SOCKET_IFNAME", "lo")})
    procs = []
    for r in range(2):
        e = dict(env)
        e.update({"RANK": str(r), "LOCAL_RANK": str(r)})
        procs.append(subprocess.Popen([sys.executable, worker], env=e,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT, cwd=ROOT))
    outs = []
    for p in procs:
        try:
            out, _ = p.communicate(timeout=600)
        except subprocess.TimeoutExpired:
            p.kill()
            out, _ = p.communicate()
        outs.append(out.decode(errors="replace"))
    assert all(p.returncode == 0 for p in procs), outs[-1][-3000:]
    assert any("DIST_ALL_OK" in o for o in outs), outs[0][-2000:]


@pytest.mark.gpu
def test_gloo_on_gpu_gmg_repl_tail():
    """gmg.py at ws=2 on one GPU (gloo wire): the replicated coarse tail
    runs hipGraph-captured with device tensors on both ranks, with the
    fine level distributed (repl_threshold forces a mid-hierarchy cut)."""
    if _ngpus() < 1:
        pytest.skip("needs a GPU")
    argv = [sys.executable, os.path.join(ROOT, "examples", "gmg.py"),
            "-N", "255", "-repl_threshold", "20000", "-maxiter", "50"]
    env = dict(os.environ)
    env.update({"MASTER_ADDR": "127.0.0.1", "MASTER_PORT": "29747",
                "WORLD_SIZE": "2", "SPARSE_DIST_BACKEND": "gloo",
                "GLOO_SOCKET_IFNAME": env.get("GLOO_SOCKET_IFNAME", "lo")})
    procs = []
    for r in range(2):
        e = dict(env)
        e.update({"RANK": str(r), "LOCAL_RANK": str(r)})
        procs.append(subprocess.Popen(argv, env=e, stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT, cwd=ROOT))
    outs = []
    for p in procs:
        out, _ = p.communicate(timeout=600)
        outs.append(out.decode(errors="replace"))
        assert p.returncode == 0, outs[-1][-2500:]
    assert "info=0" in outs[0], outs[0][-500:]


@pytest.mark.gpu
def test_gloo_on_gpu_bench_ws2():
    """bench.py at ws=2 with both ranks on one GPU (gloo wire): the exact
    multi-rank bench code path (DIA interior/boundary split, halo
    exchange, max-over-ranks timing) with device tensors — guards the
    first contact with a real 8-GPU node."""
    if _ngpus() < 1:
        pytest.skip("needs a GPU")
    import json

    argv = [sys.executable, os.path.join(ROOT, "bench.py"), "--gpus", "2",
            "--steps", "5", "--warmup", "2", "--nx", "1024", "--weak",
            "--no-matched"]
    env = dict(os.environ)
    env.update({"MASTER_ADDR": "127.0.0.1", "MASTER_PORT": "29748",
                "WORLD_SIZE": "2", "SPARSE_DIST_BACKEND": "gloo",
                "GLOO_SOCKET_IFNAME": env.get("GLOO_SOCKET_IFNAME", "lo")})
    procs = []
    for r in range(2):
        e = dict(env)
        e.update({"RANK": str(r), "LOCAL_RANK": str(r)})
        procs.append(subprocess.Popen(argv, env=e, stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT, cwd=ROOT))
    outs = []
    for p in procs:
        out, _ = p.communicate(timeout=600)
        outs.append(out.decode(errors="replace"))
        assert p.returncode == 0, outs[-1][-2500:]
    line = [ln for ln in outs[0].splitlines() if ln.startswith("{")]
    assert len(line) == 1, outs[0][-800:]
    rec = json.loads(line[0])
    assert rec["n_gpus"] == 2 and rec["value"] > 0


@pytest.mark.gpu
@pytest.mark.parametrize("nproc", [2, 8])
def test_rccl_battery(nproc):
    if _ngpus() < nproc:
        pytest.skip(f"needs {nproc} GPUs, have {_ngpus()}")
    worker = os.path.join(ROOT, "tests", "dist_worker.py")
    rcs, outs = _launch_spmd([sys.executable, worker], nproc, 29750 + nproc)
    assert all(rc == 0 for rc in rcs), outs[-1][-3000:]
    assert any("DIST_ALL_OK" in o for o in outs), outs[0][-2000:]


@pytest.mark.gpu
def test_rccl_bench_weak():
    """bench.py --gpus N --weak over RCCL: the exact command shape the
    driver's SCALE run uses (small grid; contract + correctness only)."""
    n = min(_ngpus(), 4)
    if n < 2:
        pytest.skip(f"needs 2+ GPUs, have {_ngpus()}")
    argv = [sys.executable, os.path.join(ROOT, "bench.py"), "--gpus", str(n),
            "--steps", "20", "--warmup", "5", "--nx", "2048", "--weak",
            "--no-matched"]
    rcs, outs = _launch_spmd(argv, n, 29790)
    assert all(rc == 0 for rc in rcs), outs[0][-3000:]
    line = [ln for ln in outs[0].splitlines() if ln.startswith("{")]
    assert len(line) == 1, outs[0][-800:]
    rec = json.loads(line[0])
    assert rec["n_gpus"] == n and rec["scaling"] == "weak"
    assert rec["value"] > 0


@pytest.mark.gpu
def test_rccl_pde_example():
    """examples/pde.py at ws=2 on RCCL — the north-star end-to-end path."""
    if _ngpus() < 2:
        pytest.skip(f"needs 2 GPUs, have {_ngpus()}")
    argv = [sys.executable, os.path.join(ROOT, "examples", "pde.py"),
            "-nx", "1000", "-ny", "1000", "-throughput", "-max_iter", "100"]
    rcs, outs = _launch_spmd(argv, 2, 29795)
    assert all(rc == 0 for rc in rcs), outs[0][-3000:]
    assert "residual norm" in outs[0], outs[0][-800:]
