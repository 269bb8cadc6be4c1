"""The x-gather plan: the explicit MI355X replacement for the reference's
image partitions.

Reference parity: MinMaxImagePartition (sparse/partition.py:139-208) — each
row slab of a CSR matrix needs only x[lo:hi) where [lo,hi] is the min/max of
its column indices; the runtime there turns that into an implicit gather.
Here the plan is computed once per (matrix structure, operand partition),
cached on the matrix, and executed as one RCCL/xGMI alltoallv of the
overlapping x slices.  (settings.precise_images selects an exact-index plan
in the reference; the window plan is what its default build uses.)
"""
from __future__ import annotations

import torch

from . import comm
from .partition import RowPartition


class WindowGatherPlan:
    def __init__(self, lo: int, hi: int, xpart: RowPartition, group=None):
        self.lo = int(lo)
        self.hi = int(hi)
        self.xpart = xpart
        self.group = group
        ws = comm.world_size(group)
        me = comm.rank(group)
        if ws == 1:
            self.windows = [(self.lo, self.hi)]
        else:
            w = torch.tensor([self.lo, self.hi], dtype=torch.int64)
            outs = [torch.zeros(2, dtype=torch.int64) for _ in range(ws)]
            import torch.distributed as dist

            if dist.get_backend(group) == "nccl":
                from ..runtime import runtime

                w = w.to(runtime().device)
                outs = [o.to(runtime().device) for o in outs]
            dist.all_gather(outs, w, group=group)
            self.windows = [(int(o[0].item()), int(o[1].item())) for o in outs]
        # slices of MY slab that each peer needs
        s0, s1 = xpart.start(me), xpart.stop(me)
        self.send_ranges = []
        for p in range(ws):
            plo, phi = self.windows[p]
            a, b = max(plo, s0), min(phi, s1)
            self.send_ranges.append((a - s0, max(a, b) - s0))
        # what I receive from each peer (ascending in p => contiguous window)
        self.recv_counts = []
        for p in range(ws):
            a, b = max(self.lo, xpart.start(p)), min(self.hi, xpart.stop(p))
            self.recv_counts.append(max(0, b - a))
        assert sum(self.recv_counts) == self.hi - self.lo
        self._ctx_cache = {}

    def _halo_ctx(self, xlocal: torch.Tensor):
        """Persistent halo-exchange context for a (repeatedly updated
        in-place) operand tensor: send views into x, one contiguous recv
        buffer per side with per-peer slices — zero allocations and no
        concatenation per exchange."""
        key = (xlocal.data_ptr(), tuple(xlocal.shape), xlocal.dtype)
        ctx = self._ctx_cache.get(key)
        if ctx is not None:
            return ctx
        ws = comm.world_size(self.group)
        me = comm.rank(self.group)
        xs, xe = self.xpart.start(me), self.xpart.stop(me)
        own_a = max(self.lo, xs)
        own_b = max(own_a, min(self.hi, xe))  # clamp: window may miss my slab
        send_views = []
        for p in range(ws):
            if p == me:
                send_views.append(None)
                continue
            a, b = self.send_ranges[p]
            send_views.append(xlocal[a:b] if b > a else None)
        n_lo = sum(self.recv_counts[p] for p in range(me))
        n_hi = sum(self.recv_counts[p] for p in range(me + 1, ws))
        hlo = torch.empty(n_lo, dtype=xlocal.dtype, device=xlocal.device)
        hhi = torch.empty(n_hi, dtype=xlocal.dtype, device=xlocal.device)
        recv_views = []
        off = 0
        for p in range(ws):
            c = self.recv_counts[p]
            if p == me or c == 0:
                recv_views.append(None)
            elif p < me:
                recv_views.append(hlo[off: off + c])
            else:
                recv_views.append(hhi[off - n_lo: off - n_lo + c])
            if p != me:
                off += c
            else:
                off = n_lo  # own piece not received; hhi offsets start at 0
        own = xlocal[own_a - xs: own_b - xs]
        # device wire path: prebuild the P2POp descriptors once — at ws=8
        # the per-call Python loop + P2POp construction is measurable
        # against sub-millisecond kernels (P2POps are reusable holders)
        import torch.distributed as dist

        p2p_dev = []
        if not xlocal.is_cuda or comm._nccl(self.group):
            for p in range(ws):
                sv = send_views[p]
                if sv is not None and sv.numel():
                    p2p_dev.append(dist.P2POp(
                        dist.isend,
                        sv if sv.is_contiguous() else sv.contiguous(),
                        p, group=self.group))
                rv = recv_views[p]
                if rv is not None:
                    p2p_dev.append(dist.P2POp(dist.irecv, rv, p,
                                              group=self.group))
        ctx = (send_views, recv_views, hlo, own, hhi, p2p_dev)
        if len(self._ctx_cache) >= 8:
            # contexts hold views (keep operand storage alive): bound them
            self._ctx_cache.clear()
        self._ctx_cache[key] = ctx
        return ctx

    def gather_halos_begin(self, xlocal: torch.Tensor):
        """Start the halo exchange without waiting: returns an opaque
        handle for gather_halos_end.  Interior rows (which never read the
        halo pieces) can compute while the exchange is in flight — the
        MI355X realization of the reference's Legion-overlapped gathers.
        The pieces in the handle are valid for interior use immediately
        (halo buffers unfilled until _end)."""
        ws = comm.world_size(self.group)
        if ws == 1:
            return ("done", (xlocal[:0], xlocal[self.lo: self.hi],
                             xlocal[:0]))
        import torch.distributed as dist

        send_views, recv_views, hlo, own, hhi, p2p_dev = self._halo_ctx(xlocal)
        nothing_to_exchange = (
            not any(v is not None and v.numel() for v in send_views)
            and not any(v is not None for v in recv_views))
        if p2p_dev or nothing_to_exchange:
            # device/host-native wire: prebuilt descriptors, zero per-call
            # Python construction (p2p_dev is empty only when there is
            # nothing to exchange or the wire needs host staging below)
            reqs = dist.batch_isend_irecv(p2p_dev) if p2p_dev else []
            return ("pending", reqs, {}, recv_views, (hlo, own, hhi))
        # gloo cannot move CUDA tensors: stage the halos through the host
        # (the several-ranks-per-GPU battery configuration; RCCL runs the
        # zero-copy device path above)
        p2p = []
        host_recv = {}
        for p in range(ws):
            if send_views[p] is not None and send_views[p].numel():
                sv = send_views[p]
                if not sv.is_contiguous():
                    sv = sv.contiguous()
                p2p.append(dist.P2POp(dist.isend, sv.cpu(), p, group=self.group))
            if recv_views[p] is not None:
                rv = torch.empty(recv_views[p].shape, dtype=recv_views[p].dtype)
                host_recv[p] = rv
                p2p.append(dist.P2POp(dist.irecv, rv, p, group=self.group))
        reqs = dist.batch_isend_irecv(p2p) if p2p else []
        return ("pending", reqs, host_recv, recv_views, (hlo, own, hhi))

    @staticmethod
    def handle_pieces(handle):
        """The (hlo, own, hhi) views of an in-flight handle — valid for
        INTERIOR rows only until gather_halos_end returns."""
        return handle[1] if handle[0] == "done" else handle[4]

    def gather_halos_end(self, handle):
        """Complete an exchange started by gather_halos_begin; returns the
        (halo_lo, own_view, halo_hi) pieces with halos filled."""
        if handle[0] == "done":
            return handle[1]
        _, reqs, host_recv, recv_views, pieces = handle
        for req in reqs:
            req.wait()
        for p, rv in host_recv.items():
            recv_views[p].copy_(rv)
        return pieces

    def gather_halos(self, xlocal: torch.Tensor):
        """Exchange ONLY the halo pieces; my own slab portion is used in
        place (no self-copy through the collective).  Returns
        (halo_lo, own_view, halo_hi)."""
        return self.gather_halos_end(self.gather_halos_begin(xlocal))

    def gather(self, xlocal: torch.Tensor) -> torch.Tensor:
        """Return the window x[lo:hi) (dim 0 slices; works for 1-D and 2-D)."""
        ws = comm.world_size(self.group)
        me = comm.rank(self.group)
        if ws == 1:
            if self.lo == 0 and self.hi == xlocal.shape[0]:
                return xlocal
            return xlocal[self.lo: self.hi]
        tail = xlocal.shape[1:]
        k = 1
        for t in tail:
            k *= t
        send = []
        for p in range(ws):
            a, b = self.send_ranges[p]
            send.append(xlocal[a:b].reshape(-1))
        rc = [c * k for c in self.recv_counts]
        recv = comm.all_to_all_v(send, group=self.group, recv_counts=rc)
        out = torch.cat([r for r in recv], dim=0)
        return out.reshape(self.hi - self.lo, *tail) if tail else out


class PreciseGatherPlan:
    """Exact-index gather plan (reference precise images,
    settings.py:23-33 / MinMaxImagePartition's precise alternative): the
    operand slice communicated per rank is exactly the set of distinct
    column indices its slab touches, not the min/max window.  Pays off for
    scattered matrices at world size > 1 where the window spans nearly all
    of x but only a few columns are read.

    Interface-compatible with WindowGatherPlan's gather() path: gather()
    returns a compact array of the needed values (sorted by global column),
    lo == 0 and hi == len(cols), and remap(indices) rewrites a local index
    tensor into positions in that compact array."""

    def __init__(self, indices: torch.Tensor, xpart: RowPartition, group=None):
        self.xpart = xpart
        self.group = group
        ws = comm.world_size(group)
        me = comm.rank(group)
        # distinct needed global columns, ascending (=> grouped by owner)
        self.cols = torch.unique(indices.to(torch.int64))
        self.lo = 0
        self.hi = int(self.cols.numel())
        starts = torch.tensor(xpart.starts, dtype=torch.int64,
                              device=self.cols.device)
        cuts = torch.searchsorted(self.cols, starts)
        self._req_counts = [int(cuts[p + 1] - cuts[p]) for p in range(ws)]
        if ws == 1:
            self._send_idx = [self.cols]
            self._local_cols = self.cols
            self._remap_cache = None
            return
        # exchange the request lists: owners learn which of their entries
        # each peer needs
        reqs = [self.cols[cuts[p]: cuts[p + 1]].contiguous() for p in range(ws)]
        got = comm.all_to_all_v(reqs, group=group)
        s0 = xpart.start(me)
        self._send_idx = [g - s0 for g in got]  # my-slab-local gather indices
        self._remap_cache = None

    def remap(self, indices: torch.Tensor) -> torch.Tensor:
        """Rewrite global column indices into compact positions (cached)."""
        if self._remap_cache is None:
            self._remap_cache = torch.searchsorted(
                self.cols, indices.to(torch.int64)).to(indices.dtype)
        return self._remap_cache

    def gather(self, xlocal: torch.Tensor) -> torch.Tensor:
        ws = comm.world_size(self.group)
        if ws == 1:
            return xlocal[self.cols]
        tail = xlocal.shape[1:]
        k = 1
        for t in tail:
            k *= t
        send = [(xlocal[idx] if idx.numel() else xlocal[:0]).reshape(-1)
                for idx in self._send_idx]
        recv = comm.all_to_all_v(send, group=self.group,
                                 recv_counts=[c * k for c in self._req_counts])
        out = torch.cat(recv, dim=0)
        return out.reshape(self.hi, *tail) if tail else out


class ColBlockGatherPlan:
    """Gather the COLUMN block D[:, lo:hi) of a dim-0-partitioned 2-D
    operand — the reference's MinMaxImage on projection dim 1 for the
    SDDMM D operand (csr.py:1244-1312, sddmm.cu:25-85).  Per-rank wire
    bytes = own_rows x my_window (each owner slices its rows to each
    peer's window), not rows x n as a full gather would move."""

    def __init__(self, lo: int, hi: int, dpart: RowPartition, group=None):
        self.lo = int(lo)
        self.hi = int(hi)
        self.dpart = dpart
        self.group = group
        ws = comm.world_size(group)
        if ws == 1:
            self.windows = [(self.lo, self.hi)]
            return
        w = torch.tensor([self.lo, self.hi], dtype=torch.int64)
        outs = [torch.zeros(2, dtype=torch.int64) for _ in range(ws)]
        import torch.distributed as dist

        if dist.get_backend(group) == "nccl":
            from ..runtime import runtime

            w = w.to(runtime().device)
            outs = [o.to(runtime().device) for o in outs]
        dist.all_gather(outs, w, group=group)
        self.windows = [(int(o[0].item()), int(o[1].item())) for o in outs]

    def gather(self, dlocal: torch.Tensor) -> torch.Tensor:
        """dlocal: my (rows_p, n) slab; returns the full-height column
        block (rows, hi-lo), contiguous."""
        ws = comm.world_size(self.group)
        me = comm.rank(self.group)
        if ws == 1:
            return dlocal[:, self.lo: self.hi].contiguous()
        send = []
        for p in range(ws):
            plo, phi = self.windows[p]
            send.append(dlocal[:, plo:phi].contiguous().reshape(-1))
        w = self.hi - self.lo
        rc = [self.dpart.count(p) * w for p in range(ws)]
        recv = comm.all_to_all_v(send, group=self.group, recv_counts=rc)
        pieces = [recv[p].reshape(self.dpart.count(p), w) for p in range(ws)]
        return torch.cat(pieces, dim=0)


class ReduceScatterPlan:
    """Inverse of the window gather: each rank holds partial contributions to
    y[lo:hi); owners receive and sum them.

    Reference parity: the ADD reductions of col-split SpMV / CSC SpMV
    (csr.py:869-927, csc/spmv.cu:60-75) that Legion performs implicitly.
    """

    def __init__(self, lo: int, hi: int, ypart: RowPartition, group=None):
        self.fwd = WindowGatherPlan(lo, hi, ypart, group)

    def scatter_add(self, partial: torch.Tensor, ylocal: torch.Tensor, beta: float = 1.0) -> torch.Tensor:
        """partial covers [lo,hi); add into ylocal (owner slabs)."""
        ws = comm.world_size(self.fwd.group)
        me = comm.rank(self.fwd.group)
        if beta == 0.0:
            ylocal.zero_()
        if ws == 1:
            ylocal[self.fwd.lo: self.fwd.hi] += partial
            return ylocal
        # send partial pieces to owners: piece for peer p is the overlap of my
        # window with p's slab — the same recv_counts layout, reversed.
        xpart = self.fwd.xpart
        tail_k = 1
        for t in partial.shape[1:]:
            tail_k *= t
        send = []
        off = 0
        for p in range(ws):
            c = self.fwd.recv_counts[p]
            send.append(partial[off: off + c].reshape(-1))
            off += c
        rc = [(b - a) * tail_k for a, b in self.fwd.send_ranges]
        recv = comm.all_to_all_v(send, group=self.fwd.group, recv_counts=rc)
        s0 = xpart.start(me)
        tail = ylocal.shape[1:]
        for p in range(ws):
            a, b = self.fwd.send_ranges[p]  # my-slab-local target range for peer p's piece
            if b > a:
                ylocal[a:b] += recv[p].reshape(b - a, *tail)
        return ylocal
