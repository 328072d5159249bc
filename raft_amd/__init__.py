"""raft_amd — MI355X-native ML/data-mining primitives framework.

A brand-new AMD CDNA4 (gfx950) implementation of the capability surface of
rapidsai/raft: dense linear algebra, matrix ops and k-selection,
sparse ops and solvers, random generation, statistics/metrics, pairwise
distances, fused L2-NN, k-means, and a distributed comms layer over RCCL/xGMI.

Design (MI355X-first, not a port):
  * PyTorch-ROCm tensors are the array substrate (HBM3E-resident, dlpack interop).
  * Hot ops are hand-written HIP/CDNA4 kernels (MFMA tiles, LDS staging, wave64)
    compiled for gfx950 into the in-tree extension ``raft_amd._C``.
  * Plain library GEMMs go through rocBLAS/hipBLASLt (via torch / our wrappers).
  * Multi-GPU scaling is one process per GPU with torch.distributed over RCCL.
  * CPU tensors run a pure-PyTorch reference path (the numerics oracle used by
    the test suite); GPU tensors require the native extension for hot ops and
    fail loudly if it is missing.
"""

__version__ = "0.1.0"

from . import core
from . import utils
from . import linalg
from . import matrix
from . import random
from . import stats
from . import distance
from . import neighbors
from . import cluster
from . import sparse
from . import solver
from . import spectral
from . import label
from . import comms

__all__ = [
    "core", "utils", "linalg", "matrix", "random", "stats", "distance",
    "neighbors", "cluster", "sparse", "solver", "spectral", "label", "comms",
    "__version__",
]
