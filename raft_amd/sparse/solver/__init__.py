from .lanczos import lanczos_min_eigenpairs, eigsh, LanczosConfig
from .mst import mst
from .randomized_svd import randomized_svds
from .linear_operator import LinearOperator, csr_operator

__all__ = ["lanczos_min_eigenpairs", "eigsh", "LanczosConfig", "mst",
           "randomized_svds", "LinearOperator", "csr_operator"]
