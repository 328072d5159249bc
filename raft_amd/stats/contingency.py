"""Contingency matrix (reference: raft/stats/contingencyMatrix.cuh — binned
atomics with label-range reduction)."""
from __future__ import annotations

import torch


def contingency_matrix(labels_a: torch.Tensor, labels_b: torch.Tensor,
                       n_classes_a: int | None = None,
                       n_classes_b: int | None = None) -> torch.Tensor:
    """Label contingency table over the label RANGES (reference)."""
    a = labels_a.to(torch.int64)
    b = labels_b.to(torch.int64)
    amin, bmin = int(a.min()), int(b.min())
    a = a - amin
    b = b - bmin
    na = n_classes_a or int(a.max().item()) + 1
    nb = n_classes_b or int(b.max().item()) + 1
    flat = a * nb + b
    return torch.bincount(flat, minlength=na * nb).reshape(na, nb)
