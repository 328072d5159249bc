"""Per-row key-value sort (reference: raft/matrix/detail/columnWiseSort.cuh —
in-block cub BlockRadixSort / DeviceSegmentedRadixSort; on ROCm the segmented
sort is torch.sort over dim 1, which lowers to rocPRIM segmented radix sort).
"""
from __future__ import annotations

import torch


def col_wise_sort(keys: torch.Tensor, values: torch.Tensor | None = None, descending: bool = False):
    """Sort each row of `keys`; permute `values` identically.

    (The reference's name refers to sorting the columns *within* each row.)
    """
    sorted_keys, order = torch.sort(keys, dim=1, descending=descending)
    if values is None:
        return sorted_keys, order
    return sorted_keys, torch.gather(values, 1, order)
