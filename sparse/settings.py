"""Runtime settings (reference sparse/settings.py:22-36).

precise_images: when True (and world size > 1), SpMV/SpMM gather plans
communicate the exact set of distinct columns each rank's slab touches
(PreciseGatherPlan) instead of the min/max window — the reference's
LEGATE_SPARSE_PRECISE_IMAGES (settings.py:23-33).  The window plan is the
default, as in the reference build; banded matrices keep the window plan
regardless (their windows are already tight and feed the DIA/ELL kernels).
"""
from __future__ import annotations

import os


class Settings:
    def __init__(self) -> None:
        self.precise_images = os.environ.get(
            "SPARSE_PRECISE_IMAGES",
            os.environ.get("LEGATE_SPARSE_PRECISE_IMAGES", "0"),
        ) not in ("0", "", "false", "False")
        # cap on the number of ranks that own data (reference
        # LEGATE_SPARSE_NUM_PROCS, runtime.py:61-63); the remaining ranks
        # participate in collectives with empty slabs
        self.num_procs = int(os.environ.get("SPARSE_NUM_PROCS", "0")) or None


settings = Settings()
