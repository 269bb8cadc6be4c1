"""Multi-process distributed tests (gloo backend, CPU — runs in CI).

The same code paths carry the 8-GPU RCCL runs; SURVEY §4's "distributed
testing via more processors" obligation.
"""
import os
import subprocess
import sys

import pytest


@pytest.mark.parametrize("nproc", [2, 4])
def test_distributed_battery(nproc):
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
