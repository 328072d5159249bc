"""Classification metrics (reference: raft/stats/accuracy.cuh)."""
from __future__ import annotations

import torch


def accuracy_score(y_true: torch.Tensor, y_pred: torch.Tensor) -> float:
    """Fraction of matching labels (reference stats::accuracy)."""
    return float((y_true == y_pred).double().mean())
