"""device_ndarray — the pylibraft array type, MI355X-native.

Reference parity: python/pylibraft/pylibraft/common/device_ndarray.py:10-157
(a CAI array over an rmm DeviceBuffer). Here the storage is a torch tensor in
HBM3E; interop is via DLPack (the ROCm-native zero-copy protocol) and, on GPU,
``__cuda_array_interface__`` (which torch exposes for HIP memory under ROCm).
"""
from __future__ import annotations

import numpy as np
import torch

_NP_TO_TORCH = {
    np.dtype("float16"): torch.float16,
    np.dtype("float32"): torch.float32,
    np.dtype("float64"): torch.float64,
    np.dtype("int8"): torch.int8,
    np.dtype("uint8"): torch.uint8,
    np.dtype("int16"): torch.int16,
    np.dtype("int32"): torch.int32,
    np.dtype("int64"): torch.int64,
    np.dtype("bool"): torch.bool,
}
_TORCH_TO_NP = {v: k for k, v in _NP_TO_TORCH.items()}


class device_ndarray:
    """Lightweight device array: shape/dtype/strides over a torch tensor."""

    def __init__(self, np_or_tensor):
        if isinstance(np_or_tensor, torch.Tensor):
            self._tensor = np_or_tensor
        elif isinstance(np_or_tensor, np.ndarray):
            dev = "cuda" if torch.cuda.is_available() else "cpu"
            self._tensor = torch.from_numpy(np.ascontiguousarray(np_or_tensor)).to(dev)
        else:
            raise TypeError(f"expected numpy array or torch tensor, got {type(np_or_tensor)}")

    # -- factories ----------------------------------------------------------
    @classmethod
    def empty(cls, shape, dtype=np.float32, order="C", device=None):
        tdt = _NP_TO_TORCH[np.dtype(dtype)]
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        t = torch.empty(tuple(int(s) for s in np.atleast_1d(shape)), dtype=tdt, device=device)
        if order == "F" and t.dim() == 2:
            t = t.t().contiguous().t()
        return cls(t)

    @classmethod
    def zeros(cls, shape, dtype=np.float32, device=None):
        out = cls.empty(shape, dtype=dtype, device=device)
        out._tensor.zero_()
        return out

    # -- properties ---------------------------------------------------------
    @property
    def torch(self) -> torch.Tensor:
        return self._tensor

    @property
    def shape(self):
        return tuple(self._tensor.shape)

    @property
    def dtype(self):
        return _TORCH_TO_NP[self._tensor.dtype]

    @property
    def strides(self):
        return tuple(s * self._tensor.element_size() for s in self._tensor.stride())

    @property
    def c_contiguous(self) -> bool:
        return self._tensor.is_contiguous()

    @property
    def f_contiguous(self) -> bool:
        return self._tensor.t().is_contiguous() if self._tensor.dim() == 2 else False

    @property
    def __cuda_array_interface__(self):
        return self._tensor.__cuda_array_interface__

    def __dlpack__(self, stream=None):
        return self._tensor.__dlpack__(stream=stream)

    def __dlpack_device__(self):
        return self._tensor.__dlpack_device__()

    # -- conversion ---------------------------------------------------------
    def copy_to_host(self) -> np.ndarray:
        return self._tensor.detach().cpu().numpy()

    def __array__(self, dtype=None):
        a = self.copy_to_host()
        return a.astype(dtype) if dtype is not None else a

    def __len__(self):
        return self._tensor.shape[0]

    def __repr__(self):  # pragma: no cover
        return f"device_ndarray(shape={self.shape}, dtype={self.dtype}, device={self._tensor.device})"
