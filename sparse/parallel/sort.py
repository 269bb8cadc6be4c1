"""Distributed samplesort of key/payload tuples.

Reference parity: SORT_BY_KEY (src/sparse/sort/sort.cu:124-379): local sort,
sample extraction (sort.cu:27-73), all-gather of samples (:163-168),
splitter selection + local split search (:75-122,171-197), all-to-allv of
the payload (:257-322, with zero-size guards), and a final local merge
(:329-377 — here a single local re-sort, same complexity class on GPU).

Used when the target distribution is unknown (e.g. nnz-balanced COO
repartitioning); the owner-map shuffle in shuffle.py covers the known-owner
cases (COO->CSR with an equal row tiling).
"""
from __future__ import annotations

import torch

from . import comm


def samplesort(key: torch.Tensor, *payload, oversample: int = 32, group=None):
    """Globally sort (key, *payload) by key across ranks.

    Returns the local chunk of the globally sorted sequence (rank r holds
    keys <= rank r+1's keys); chunk sizes are approximately balanced.
    """
    ws = comm.world_size(group)
    order = torch.argsort(key, stable=True)
    key = key[order]
    payload = tuple(p[order] for p in payload)
    if ws == 1:
        return (key, *payload)
    n = key.numel()
    # sample extraction: ws*oversample evenly spaced local samples
    ns = ws * oversample
    if n > 0:
        pos = (torch.arange(ns, device=key.device, dtype=torch.float64) + 0.5) * n / ns
        samples = key[pos.long().clamp(max=n - 1)]
    else:
        samples = key[:0]
    # all-gather samples (padded equal-size gather handles empty ranks)
    counts = torch.zeros(ws, dtype=torch.int64)
    counts[comm.rank(group)] = samples.numel()
    comm.all_reduce_(counts, group=group)
    all_samples = comm.all_gather_rows(samples, [int(c) for c in counts],
                                       group=group)
    all_samples, _ = torch.sort(all_samples)
    # ws-1 splitters
    m = all_samples.numel()
    if m == 0:
        splitters = key[:0]
    else:
        idx = (torch.arange(1, ws, device=all_samples.device) * m) // ws
        splitters = all_samples[idx]
    # local split positions; send chunk p to rank p
    bounds = torch.searchsorted(key, splitters.to(key.dtype))
    bounds = torch.cat([torch.zeros(1, dtype=torch.int64, device=key.device),
                        bounds.to(torch.int64),
                        torch.tensor([n], dtype=torch.int64, device=key.device)])
    send_k = [key[bounds[p]: bounds[p + 1]] for p in range(ws)]
    recv_k = comm.all_to_all_v(send_k, group=group)
    out_payload = []
    for t in payload:
        send_p = [t[bounds[p]: bounds[p + 1]] for p in range(ws)]
        out_payload.append(torch.cat(comm.all_to_all_v(send_p, group=group)))
    rk = torch.cat(recv_k)
    # final local sort (received runs are pre-sorted; a re-sort is the
    # GPU-simple equivalent of the reference's log(p) merge rounds)
    order = torch.argsort(rk, stable=True)
    return (rk[order], *[p[order] for p in out_payload])
