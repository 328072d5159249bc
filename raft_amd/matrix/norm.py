"""matrix::l2_norm wrapper over linalg (reference: raft/matrix/norm.cuh)."""
from __future__ import annotations

import torch

from raft_amd.linalg.norm import row_norm, col_norm, NormType


def l2_norm(x: torch.Tensor, along_rows: bool = True, sqrt: bool = True) -> torch.Tensor:
    """Row (or column) L2 norms (reference matrix::l2norm wrapper)."""
    fn = row_norm if along_rows else col_norm
    return fn(x, NormType.L2, sqrt=sqrt)
