"""Runtime singleton: device selection, distributed context.

This replaces the Legion/Legate runtime of the reference
(sparse/runtime.py:75-126): instead of a dynamic partitioning solver we run
SPMD — one process per GPU (torch.distributed, backend "nccl" == RCCL on
ROCm, "gloo" on CPU) — and every distributed op issues its own explicit
collectives.

Overlap model (RCCL): torch.distributed's NCCL process group launches every
collective/p2p on its OWN internal HIP stream per device; posting
`batch_isend_irecv` returns immediately, compute kernels enqueued afterwards
on the default stream run concurrently with the RCCL kernels over xGMI, and
`work.wait()` only makes the CURRENT stream wait on the comm-stream event
(no host block).  Halo/interior overlap (gather_halos_begin → interior
kernels → gather_halos_end → boundary kernels) therefore needs no extra
application-managed stream; we keep none.
"""
from __future__ import annotations

import os
from typing import Optional

import torch
import torch.distributed as dist


class Runtime:
    _instance: Optional["Runtime"] = None

    def __init__(self) -> None:
        self._initialized_dist = False
        self.rank = 0
        self.world_size = 1
        if dist.is_available() and dist.is_initialized():
            self._adopt_dist()
        elif "RANK" in os.environ and "WORLD_SIZE" in os.environ and int(os.environ["WORLD_SIZE"]) > 1:
            self._init_dist_from_env()

        self.use_gpu = torch.cuda.is_available()
        if self.use_gpu:
            local_rank = int(os.environ.get(
                "LOCAL_RANK", self.rank)) % max(1, torch.cuda.device_count())
            torch.cuda.set_device(local_rank)
            self.device = torch.device("cuda", local_rank)
        else:
            self.device = torch.device("cpu")
        self.num_gpus = torch.cuda.device_count() if self.use_gpu else 0

    # -- distributed bring-up -------------------------------------------------
    def _init_dist_from_env(self) -> None:
        # SPARSE_DIST_BACKEND overrides (e.g. gloo on a GPU box: several
        # ranks sharing one device — the distributed-battery-on-hardware
        # test configuration)
        backend = os.environ.get("SPARSE_DIST_BACKEND") or (
            "nccl" if torch.cuda.is_available() else "gloo")
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29571")
        dist.init_process_group(backend=backend)
        self._adopt_dist()
        self._initialized_dist = True
        # tear the process group down at interpreter exit: gloo/NCCL
        # destructors racing interpreter shutdown otherwise abort with
        # "terminate called without an active exception"
        import atexit

        atexit.register(self.finalize)

    def _adopt_dist(self) -> None:
        self.rank = dist.get_rank()
        self.world_size = dist.get_world_size()

    @property
    def distributed(self) -> bool:
        return self.world_size > 1

    @property
    def backend(self) -> Optional[str]:
        if dist.is_available() and dist.is_initialized():
            return dist.get_backend()
        return None

    def barrier(self) -> None:
        if self.distributed:
            dist.barrier()

    def finalize(self) -> None:
        if self._initialized_dist and dist.is_initialized():
            dist.destroy_process_group()
            self._initialized_dist = False


_runtime: Optional[Runtime] = None


def runtime() -> Runtime:
    global _runtime
    if _runtime is None:
        _runtime = Runtime()
    elif dist.is_available() and dist.is_initialized() and _runtime.world_size != dist.get_world_size():
        # torch.distributed was initialized after first use: re-sync.
        _runtime._adopt_dist()
    return _runtime


def reset_runtime() -> None:
    """Testing hook: force re-detection of the distributed context."""
    global _runtime
    _runtime = None
