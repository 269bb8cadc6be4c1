"""Build the HIP extension in-tree: python -m sparse.kernels.build

Compiles sparse/kernels/src/*.{cpp,hip} for gfx950 into
sparse/kernels/_build/sparse_hip.so (the .so travels with the repo snapshot
to GPU boxes; no JIT cache involvement).
"""
from __future__ import annotations

import glob
import os
import sys

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils import cpp_extension  # noqa: E402

_DIR = os.path.dirname(os.path.abspath(__file__))
BUILD_DIR = os.path.join(_DIR, "_build")


def build(verbose: bool = True) -> str:
    os.makedirs(BUILD_DIR, exist_ok=True)
    sources = sorted(glob.glob(os.path.join(_DIR, "src", "*.cpp"))) + sorted(
        glob.glob(os.path.join(_DIR, "src", "*.hip")))
    cpp_extension.load(
        name="sparse_hip",
        sources=sources,
        build_directory=BUILD_DIR,
        extra_cflags=["-O3"],
        extra_cuda_cflags=["-O3", "-std=c++17"],
        verbose=verbose,
        is_python_module=False,
        with_cuda=True,
    )
    so = os.path.join(BUILD_DIR, "sparse_hip.so")
    assert os.path.exists(so), f"build produced no {so}"
    return so


if __name__ == "__main__":
    print(build(verbose=True))
    sys.exit(0)
