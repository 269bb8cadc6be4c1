from . import comm  # noqa: F401
from .partition import RowPartition  # noqa: F401
