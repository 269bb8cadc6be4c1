"""Row/nnz shuffles of sparse shards: the explicit equivalents of the
reference's image-partition copies and samplesort-driven repartitioning.

- gather_csr_rows: each rank pulls rows [lo,hi) of a row-partitioned CSR
  (reference: the CompressedImage/MinMaxImage gathers feeding SpGEMM,
  csr.py:1322-1389).
- shuffle_coo_to_owner: alltoallv of COO triples to the rank owning each
  destination row (used by COO->CSR/CSC and distributed transpose; plays the
  role of SORT_BY_KEY's samplesort, sort.cu:124-379, with the owner map known
  a priori).
- repartition_csr: move a row-partitioned CSR onto a new RowPartition
  (reference balance(), base.py:198-282).
"""
from __future__ import annotations

from typing import Tuple

import torch

from . import comm
from .partition import RowPartition


def _csr_slice(indptr, indices, values, a, b):
    """Rows [a,b) of a local CSR -> (counts, indices, values)."""
    s, e = int(indptr[a].item()), int(indptr[b].item())
    counts = indptr[a + 1: b + 1] - indptr[a:b]
    return counts, indices[s:e], values[s:e]


def gather_csr_rows(indptr, indices, values, mypart: RowPartition, lo: int, hi: int,
                    group=None) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Return (indptr, indices, values) of global rows [lo,hi), assembled on
    this rank.  Every rank calls with its own [lo,hi)."""
    ws = comm.world_size(group)
    me = comm.rank(group)
    if ws == 1:
        counts, idx, vals = _csr_slice(indptr, indices, values, lo, hi)
        out_indptr = torch.zeros(hi - lo + 1, dtype=torch.int64, device=indptr.device)
        torch.cumsum(counts, 0, out=out_indptr[1:])
        return out_indptr, idx, vals
    # exchange requested windows
    import torch.distributed as dist

    dev = indices.device
    cdev = dev if dist.get_backend(group) == "nccl" else torch.device("cpu")
    w = torch.tensor([lo, hi], dtype=torch.int64, device=cdev)
    outs = [torch.zeros(2, dtype=torch.int64, device=cdev) for _ in range(ws)]
    dist.all_gather(outs, w, group=group)
    windows = [(int(o[0].item()), int(o[1].item())) for o in outs]
    s0, s1 = mypart.start(me), mypart.stop(me)
    send_counts, send_idx, send_val = [], [], []
    for p in range(ws):
        plo, phi = windows[p]
        a, b = max(plo, s0), min(phi, s1)
        if b <= a:
            z = torch.zeros(0, dtype=torch.int64, device=dev)
            send_counts.append(z)
            send_idx.append(indices[:0])
            send_val.append(values[:0])
        else:
            c, ii, vv = _csr_slice(indptr, indices, values, a - s0, b - s0)
            send_counts.append(c.to(torch.int64))
            send_idx.append(ii)
            send_val.append(vv)
    rc = comm.all_to_all_v(send_counts, group=group)
    ri = comm.all_to_all_v(send_idx, group=group)
    rv = comm.all_to_all_v(send_val, group=group)
    counts = torch.cat(rc)
    idx = torch.cat(ri)
    vals = torch.cat(rv)
    assert counts.numel() == hi - lo, (counts.numel(), lo, hi)
    out_indptr = torch.zeros(hi - lo + 1, dtype=torch.int64, device=indptr.device)
    torch.cumsum(counts, 0, out=out_indptr[1:])
    return out_indptr, idx, vals


def gather_csr_rows_precise(indptr, indices, values, mypart: RowPartition,
                            rows: torch.Tensor, group=None
                            ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Ship ONLY the requested rows: `rows` is this rank's sorted distinct
    global row-id list; returns (indptr, indices, values) over those rows
    in order.  The row analog of PreciseGatherPlan: for a
    scattered SpGEMM operand per-rank traffic is O(nnz of referenced
    rows), not O(nnz of the min/max window).  Every rank calls with its
    own request list (collective)."""
    ws = comm.world_size(group)
    me = comm.rank(group)
    dev = indices.device
    rows = rows.to(torch.int64)
    if ws == 1:
        counts = indptr[rows + 1] - indptr[rows]
        out_indptr = torch.zeros(rows.numel() + 1, dtype=torch.int64, device=dev)
        torch.cumsum(counts, 0, out=out_indptr[1:])
        tot = int(out_indptr[-1].item())
        pos = (torch.arange(tot, dtype=torch.int64, device=dev)
               - torch.repeat_interleave(out_indptr[:-1], counts)
               + torch.repeat_interleave(indptr[rows], counts))
        return out_indptr, indices[pos], values[pos]
    # 1. split my request list by owning rank (rows sorted => contiguous)
    starts = torch.tensor(mypart.starts, dtype=torch.int64, device=rows.device)
    cuts = torch.searchsorted(rows, starts)
    reqs = [rows[cuts[p]: cuts[p + 1]].contiguous() for p in range(ws)]
    # 2. owners receive the row ids each peer wants
    got = comm.all_to_all_v(reqs, group=group)
    s0 = mypart.start(me)
    send_counts, send_idx, send_val = [], [], []
    for g in got:
        if g.numel() == 0:
            send_counts.append(torch.zeros(0, dtype=torch.int64, device=dev))
            send_idx.append(indices[:0])
            send_val.append(values[:0])
            continue
        r = g.to(dev) - s0
        c = (indptr[r + 1] - indptr[r]).to(torch.int64)
        tot = int(c.sum().item())
        off = torch.zeros(r.numel(), dtype=torch.int64, device=dev)
        torch.cumsum(c[:-1], 0, out=off[1:])
        pos = (torch.arange(tot, dtype=torch.int64, device=dev)
               - torch.repeat_interleave(off, c)
               + torch.repeat_interleave(indptr[r], c))
        send_counts.append(c)
        send_idx.append(indices[pos])
        send_val.append(values[pos])
    # 3. ship counts + payloads back (row order within each peer's reply
    # matches its request order; concatenation over ascending owners
    # matches the sorted `rows` order)
    rc = comm.all_to_all_v(send_counts, group=group)
    ri = comm.all_to_all_v(send_idx, group=group)
    rv = comm.all_to_all_v(send_val, group=group)
    counts = torch.cat(rc)
    assert counts.numel() == rows.numel()
    out_indptr = torch.zeros(rows.numel() + 1, dtype=torch.int64, device=dev)
    torch.cumsum(counts, 0, out=out_indptr[1:])
    return out_indptr, torch.cat(ri), torch.cat(rv)


def shuffle_to_owner(key: torch.Tensor, part: RowPartition, *payload, group=None):
    """Send each element to the rank owning key[t] under part.  Returns the
    concatenated (key, *payload) received, unsorted across sources."""
    ws = comm.world_size(group)
    if ws == 1:
        return (key, *payload)
    starts = torch.tensor(part.starts[1:-1], dtype=torch.int64, device=key.device)
    dest = torch.bucketize(key.to(torch.int64), starts, right=True)
    order = torch.argsort(dest, stable=True)
    sorted_dest = dest[order]
    # split points per destination rank
    bounds = torch.searchsorted(sorted_dest, torch.arange(ws + 1, device=key.device))
    outs = []
    for t in (key, *payload):
        ts = t[order]
        send = [ts[bounds[p]: bounds[p + 1]] for p in range(ws)]
        outs.append(torch.cat(comm.all_to_all_v(send, group=group)))
    return tuple(outs)


def repartition_csr(indptr, indices, values, oldpart: RowPartition,
                    newpart: RowPartition, group=None):
    """Move row slabs from oldpart to newpart; returns local (indptr, indices,
    values) under newpart."""
    ws = comm.world_size(group)
    me = comm.rank(group)
    if ws == 1 or oldpart == newpart:
        return indptr, indices, values
    dev = indices.device
    s0, s1 = oldpart.start(me), oldpart.stop(me)
    send_counts, send_idx, send_val = [], [], []
    for p in range(ws):
        a, b = max(newpart.start(p), s0), min(newpart.stop(p), s1)
        if b <= a:
            send_counts.append(torch.zeros(0, dtype=torch.int64, device=dev))
            send_idx.append(indices[:0])
            send_val.append(values[:0])
        else:
            c, ii, vv = _csr_slice(indptr, indices, values, a - s0, b - s0)
            send_counts.append(c.to(torch.int64))
            send_idx.append(ii)
            send_val.append(vv)
    rc = comm.all_to_all_v(send_counts, group=group)
    ri = comm.all_to_all_v(send_idx, group=group)
    rv = comm.all_to_all_v(send_val, group=group)
    counts = torch.cat(rc)
    m = newpart.count(me)
    assert counts.numel() == m
    out_indptr = torch.zeros(m + 1, dtype=torch.int64, device=dev)
    torch.cumsum(counts, 0, out=out_indptr[1:])
    return out_indptr, torch.cat(ri), torch.cat(rv)
