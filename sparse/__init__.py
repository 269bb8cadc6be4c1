"""sparse: an MI355X-native distributed sparse linear algebra package.

Drop-in scipy.sparse-style API (reference: nv-legate/legate.sparse,
sparse/__init__.py) backed by PyTorch-ROCm tensors, hand-written HIP/CDNA4
(gfx950) kernels and RCCL collectives over xGMI — one process per GPU.
"""
from .module import *  # noqa: F401,F403
from .coverage import clone_module, track_provenance  # noqa: F401

from .csr import csr_array, csr_matrix  # noqa: F401
from .csc import csc_array, csc_matrix  # noqa: F401
from .coo import coo_array, coo_matrix  # noqa: F401
from .dia import dia_array, dia_matrix  # noqa: F401

from .darray import DistArray, asdistarray  # noqa: F401
from . import darray  # noqa: F401
from . import io  # noqa: F401
from . import linalg  # noqa: F401
from .runtime import runtime  # noqa: F401

import scipy.sparse as _sp

clone_module(_sp, globals())

del clone_module
del _sp

__version__ = "0.1.0"
