"""minivite_amd — MI355X-native distributed Louvain (miniVite drop-in).

The product path: host C++ graph plumbing + hand-written gfx950 HIP kernels
+ RCCL over xGMI, behind the C-ABI in include/minivite_hip.h. This package
is a thin ctypes mirror of that ABI. There is NO CPU fallback: creating an
engine without a GPU fails loudly.
"""
from .api import (  # noqa: F401
    Graph,
    Engine,
    LoopbackSession,
    comm_id,
    lib,
)

__version__ = "0.1.0"
