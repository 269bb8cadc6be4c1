#!/usr/bin/env python3
"""Test runner sweeping processor configurations (role of reference test.py:
21-39, which re-executes the pytest files under varying --cpus/--gpus).

python test.py              # CPU suite at 1 rank + gloo at 2 and 4 ranks
python test.py --gpus 1     # adds the GPU-marked tests
python test.py --ranks 2 4  # choose multi-process configs
"""
import argparse
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.abspath(__file__))


def run(cmd, **kw):
    print("+", " ".join(cmd), flush=True)
    return subprocess.call(cmd, cwd=ROOT, **kw)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=0)
    p.add_argument("--ranks", type=int, nargs="*", default=[2, 4])
    p.add_argument("-k", default=None)
    args = p.parse_args()

    rc = 0
    base = [sys.executable, "-m", "pytest", "tests", "-q"]
    if args.k:
        base += ["-k", args.k]
    marker = "gpu or not gpu" if args.gpus else "not gpu"
    rc |= run(base + ["-m", marker])
    # multi-process battery re-runs with the ranks requested
    for r in args.ranks:
        env = dict(os.environ)
        rc |= run([sys.executable, "-m", "pytest",
                   "tests/test_distributed.py", "-q", "-k", f"[{r}]"], env=env)
    sys.exit(rc)


if __name__ == "__main__":
    main()
