"""Row partitions.

Replaces the reference's Legion partitioning machinery (tilings, images,
preimages — sparse/partition.py) with an explicit descriptor: a global row
dimension split into one contiguous slab per rank.  The default is the equal
row tiling (reference csr.py:242-246); `balanced_from_indptr` is the
nnz-balanced repartitioning of DenseSparseBase.balance() (base.py:198-282),
done here as a host-side binary search over the global indptr.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Sequence

import numpy as np


@dataclass(frozen=True)
class RowPartition:
    """A split of range(n) into world_size contiguous slabs.

    starts has length world_size+1; rank r owns [starts[r], starts[r+1]).
    """

    n: int
    starts: tuple

    @staticmethod
    def equal(n: int, world_size: int) -> "RowPartition":
        # Equal tiles, remainder spread over the first ranks (matches the
        # reference runtime's default tiling semantics).  SPARSE_NUM_PROCS
        # caps how many ranks own rows (reference LEGATE_SPARSE_NUM_PROCS);
        # later ranks get empty slabs but still join collectives.
        from ..settings import settings

        owners = world_size
        if settings.num_procs:
            owners = max(1, min(world_size, settings.num_procs))
        base, rem = divmod(n, owners)
        starts = [0]
        for r in range(world_size):
            if r < owners:
                starts.append(starts[-1] + base + (1 if r < rem else 0))
            else:
                starts.append(starts[-1])
        return RowPartition(n, tuple(starts))

    @staticmethod
    def single(n: int) -> "RowPartition":
        return RowPartition(n, (0, n))

    @staticmethod
    def from_starts(starts: Sequence[int]) -> "RowPartition":
        return RowPartition(int(starts[-1]), tuple(int(s) for s in starts))

    @staticmethod
    def balanced_from_counts(row_nnz: np.ndarray, world_size: int) -> "RowPartition":
        """nnz-balanced row slabs: split the cumulative nnz into equal tiles
        and take the preimage onto rows (reference base.py:198-282)."""
        n = len(row_nnz)
        if world_size == 1:
            return RowPartition.single(n)
        cum = np.concatenate([[0], np.cumsum(row_nnz, dtype=np.int64)])
        total = int(cum[-1])
        starts = [0]
        for r in range(1, world_size):
            target = (total * r) // world_size
            # first row whose cumulative nnz exceeds the target; keep slabs
            # disjoint and monotone (the reference's disjointness fix-up).
            s = int(np.searchsorted(cum, target, side="left"))
            s = max(s, starts[-1])
            s = min(s, n)
            starts.append(s)
        starts.append(n)
        return RowPartition(n, tuple(starts))

    @property
    def world_size(self) -> int:
        return len(self.starts) - 1

    def start(self, rank: int) -> int:
        return self.starts[rank]

    def stop(self, rank: int) -> int:
        return self.starts[rank + 1]

    def count(self, rank: int) -> int:
        return self.starts[rank + 1] - self.starts[rank]

    def counts(self):
        return [self.count(r) for r in range(self.world_size)]

    def owner_of(self, row: int) -> int:
        import bisect

        return bisect.bisect_right(self.starts, row) - 1 if row < self.n else self.world_size - 1

    def __eq__(self, other) -> bool:
        return isinstance(other, RowPartition) and self.starts == other.starts

    def __hash__(self):
        return hash(self.starts)
