"""Runtime settings (reference sparse/settings.py:22-36).

precise_images: when True, gather plans use the exact index set instead of
the min/max window (reference LEGATE_SPARSE_PRECISE_IMAGES).  The window
plan is the default, as in the reference build.
"""
from __future__ import annotations

import os


class Settings:
    def __init__(self) -> None:
        self.precise_images = os.environ.get(
            "SPARSE_PRECISE_IMAGES",
            os.environ.get("LEGATE_SPARSE_PRECISE_IMAGES", "0"),
        ) not in ("0", "", "false", "False")
        if self.precise_images:
            import warnings

            warnings.warn(
                "SPARSE_PRECISE_IMAGES: exact-index gather plans are not "
                "implemented; min/max window plans (the reference default "
                "build's behavior) are used", UserWarning)
        # cap on the number of ranks that own data (reference
        # LEGATE_SPARSE_NUM_PROCS, runtime.py:61-63); the remaining ranks
        # participate in collectives with empty slabs
        self.num_procs = int(os.environ.get("SPARSE_NUM_PROCS", "0")) or None


settings = Settings()
