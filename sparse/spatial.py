"""spatial.cdist — euclidean pairwise distances.

Reference parity: sparse/spatial.py:33-85 + EUCLIDEAN_CDIST task
(src/sparse/spatial/euclidean_distance.cu:28-61).  XA is row-partitioned,
XB replicated per rank (the reference's 2-D grid launch covers the same
data movement on one node); GPU path is a HIP kernel.
"""
from __future__ import annotations

import torch

from .coverage import track_provenance
from .darray import DistArray, asdistarray

__all__ = ["cdist"]


@track_provenance(nested=True)
def cdist(XA, XB, metric="euclidean", out=None):
    if metric != "euclidean":
        raise NotImplementedError("only euclidean cdist is supported "
                                  "(reference spatial.py:33)")
    XA = asdistarray(XA)
    XB = asdistarray(XB)
    if XA.ndim != 2 or XB.ndim != 2 or XA.shape[1] != XB.shape[1]:
        raise ValueError("XA and XB must be 2-D with equal column count")
    Bg = XB.gather()
    if XA.local.is_cuda:
        from . import kernels

        kernels.require()
        D = torch.empty((XA.local.shape[0], Bg.shape[0]),
                        dtype=XA.local.dtype, device=XA.local.device)
        kernels.cdist(XA.local.contiguous(), Bg.contiguous(), D)
    else:
        D = torch.cdist(XA.local.double(), Bg.double()).to(XA.local.dtype)
    res = DistArray.from_local(D, XA.partition, (XA.shape[0], XB.shape[0]))
    if out is not None:
        out.local.copy_(res.local)
        return out
    return res
