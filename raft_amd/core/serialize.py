"""NumPy .npy serialization of host/device arrays.

Reference parity: raft/core/numpy_serializer.hpp:10-41 + serialize.hpp (mdspan
serialize/deserialize in NPY format, dtype header encode/parse). Device tensors
round-trip through pinned host staging; files are standard .npy so they are
directly loadable by NumPy (the reference tests exactly this property).
"""
from __future__ import annotations

import io

import numpy as np
import torch


def serialize_mdspan(fileobj, tensor: torch.Tensor) -> None:
    """Write a tensor (host or device) to a file object in .npy format."""
    arr = tensor.detach().cpu().numpy()
    np.save(fileobj, arr, allow_pickle=False)


def deserialize_mdspan(fileobj, device=None) -> torch.Tensor:
    """Read a .npy stream into a tensor on `device` (default: CPU)."""
    arr = np.load(fileobj, allow_pickle=False)
    t = torch.from_numpy(np.ascontiguousarray(arr))
    if device is not None:
        t = t.to(device)
    return t


def save_npy(path: str, tensor: torch.Tensor) -> None:
    with open(path, "wb") as f:
        serialize_mdspan(f, tensor)


def load_npy(path: str, device=None) -> torch.Tensor:
    with open(path, "rb") as f:
        return deserialize_mdspan(f, device=device)


def dumps(tensor: torch.Tensor) -> bytes:
    buf = io.BytesIO()
    serialize_mdspan(buf, tensor)
    return buf.getvalue()


def loads(data: bytes, device=None) -> torch.Tensor:
    return deserialize_mdspan(io.BytesIO(data), device=device)
