"""pyamg -> sparse (MI355X) bridge (capability parity with reference
examples/pyamg_to_legate: patcher.py + wrapper.py)."""
from .wrapper import from_pyamg, patch  # noqa: F401
