"""pylibraft-style compatibility surface.

Reference parity: python/pylibraft — a drop-in import layer so
a pylibraft user finds the familiar names:

    from raft_amd.compat import DeviceResources, device_ndarray
    from raft_amd.compat.sparse_linalg import eigsh, svds
    from raft_amd.compat.random import rmat

The implementations are raft_amd's own (torch tensors / HIP kernels); inputs
accept anything exposing __dlpack__ or __cuda_array_interface__.
"""
from __future__ import annotations

import types

import numpy as np
import torch

from raft_amd.core import DeviceResources, Handle, device_ndarray
from raft_amd.core.resources import get_resources


def _as_tensor(x) -> torch.Tensor:
    if isinstance(x, torch.Tensor):
        return x
    if isinstance(x, device_ndarray):
        return x.torch
    if hasattr(x, "__dlpack__"):
        return torch.from_dlpack(x)
    if hasattr(x, "__cuda_array_interface__"):
        return torch.as_tensor(x, device="cuda")
    return torch.as_tensor(np.asarray(x))


# -- pylibraft.sparse.linalg -------------------------------------------------

def eigsh(A, k=6, v0=None, ncv=None, maxiter=None, tol=0, seed=None):
    """pylibraft.sparse.linalg.eigsh-compatible (smallest-algebraic Lanczos).

    A: (rows, cols, vals, shape) CSR tuple, scipy CSR, raft_amd CSR, or a
    torch sparse tensor.
    """
    from raft_amd.sparse.solver.lanczos import eigsh as _eigsh
    csr = _to_csr(A)
    w, v = _eigsh(csr, k=k, ncv=ncv or 0, maxiter=maxiter or 100,
                  tol=tol or 1e-9, seed=42 if seed is None else seed,
                  v0=None if v0 is None else _as_tensor(v0))
    return device_ndarray(w), device_ndarray(v)


def svds(A, k=6, n_iter=4, seed=42):
    """Randomized sparse SVD (pylibraft.sparse.linalg.svds analog)."""
    from raft_amd.sparse.solver import randomized_svds
    u, s, v = randomized_svds(_to_csr(A), k=k, n_iter=n_iter, seed=seed)
    return device_ndarray(u), device_ndarray(s), device_ndarray(v)


def _to_csr(A):
    from raft_amd.sparse.types import CSR
    if isinstance(A, CSR):
        return A
    if isinstance(A, torch.Tensor) and A.layout == torch.sparse_csr:
        return CSR.from_torch_sparse(A)
    if hasattr(A, "indptr") and hasattr(A, "indices") and hasattr(A, "data"):
        # scipy.sparse CSR
        dev = "cuda" if torch.cuda.is_available() else "cpu"
        return CSR(torch.as_tensor(np.asarray(A.indptr), dtype=torch.int64).to(dev),
                   torch.as_tensor(np.asarray(A.indices), dtype=torch.int64).to(dev),
                   torch.as_tensor(np.asarray(A.data)).to(dev),
                   A.shape[0], A.shape[1])
    if isinstance(A, tuple) and len(A) == 4:
        indptr, indices, vals, shape = A
        return CSR(_as_tensor(indptr).long(), _as_tensor(indices).long(),
                   _as_tensor(vals), shape[0], shape[1])
    raise TypeError(f"cannot interpret {type(A)} as CSR")


# -- pylibraft.random --------------------------------------------------------

def rmat(out, theta, r_scale, c_scale, seed=12345, handle=None):
    """pylibraft.random.rmat-compatible: fills out [n_edges, 2] with edges."""
    from raft_amd.random import rmat as _rmat
    from raft_amd.random.rng import RngState
    out_t = _as_tensor(out)
    n_edges = out_t.shape[0]
    th = _as_tensor(theta).reshape(-1, 4)
    src, dst = _rmat(r_scale, c_scale, n_edges, theta=th,
                     state=RngState(seed=seed), device=out_t.device)
    out_t[:, 0] = src.to(out_t.dtype)
    out_t[:, 1] = dst.to(out_t.dtype)
    return out


# -- pylibraft.distance / matrix --------------------------------------------

def pairwise_distance(X, Y, out=None, metric="euclidean", p=2.0, handle=None):
    from raft_amd.distance import pairwise_distance as _pd
    mmap = {"euclidean": "euclidean", "l2": "euclidean",
            "sqeuclidean": "sqeuclidean", "cosine": "cosine", "l1": "l1",
            "cityblock": "l1", "manhattan": "l1", "chebyshev": "linf",
            "linf": "linf", "minkowski": "lp", "hamming": "hamming",
            "canberra": "canberra", "inner_product": "inner_product",
            "jensenshannon": "jensenshannon", "kl_divergence": "kl_divergence",
            "correlation": "correlation", "russellrao": "russelrao"}
    d = _pd(_as_tensor(X), _as_tensor(Y), metric=mmap[metric], p=p)
    if out is not None:
        _as_tensor(out).copy_(d)
        return out
    return device_ndarray(d)


def select_k(data, k, select_min=True, handle=None):
    from raft_amd.matrix import select_k as _sk
    v, i = _sk(_as_tensor(data), k, select_min=select_min)
    return device_ndarray(v), device_ndarray(i)


# namespace-style accessors mirroring pylibraft's module layout
sparse_linalg = types.SimpleNamespace(eigsh=eigsh, svds=svds)
random = types.SimpleNamespace(rmat=rmat)
common = types.SimpleNamespace(DeviceResources=DeviceResources, Handle=Handle,
                               device_ndarray=device_ndarray)

__all__ = ["DeviceResources", "Handle", "device_ndarray", "eigsh", "svds",
           "rmat", "pairwise_distance", "select_k", "sparse_linalg", "random",
           "common", "get_resources", "Stream", "auto_sync_handle",
           "set_output_as", "post_output", "interruptible"]


# -- pylibraft.common parity ----------------------------------
# Stream, auto_sync_handle, interruptible signal bridge, set_output_as.

class Stream:
    """pylibraft.common.Stream parity: thin HIP-stream handle."""

    def __init__(self, device=None):
        self._s = (torch.cuda.Stream(device=device)
                   if torch.cuda.is_available() else None)

    def sync(self):
        if self._s is not None:
            self._s.synchronize()

    @property
    def torch_stream(self):
        return self._s


def auto_sync_handle(fn):
    """Decorator parity (pylibraft handle.pyx auto_sync_handle): calls with a
    default handle when none given and syncs it afterwards."""
    import functools
    import inspect

    @functools.wraps(fn)
    def wrapper(*args, **kwargs):
        handle = kwargs.get("handle")
        made = False
        if handle is None:
            kwargs["handle"] = get_resources()
            made = True
        out = fn(*args, **kwargs)
        if made and torch.cuda.is_available():
            torch.cuda.synchronize()
        return out

    return wrapper


_OUTPUT_AS = "torch"


def set_output_as(kind) -> None:
    """pylibraft.config.set_output_as parity: 'torch' | 'array' (numpy) |
    callable applied to outputs of `post_output`."""
    global _OUTPUT_AS
    assert kind in ("torch", "array", "cupy") or callable(kind)
    _OUTPUT_AS = kind


def post_output(t: torch.Tensor):
    """Convert an output tensor per set_output_as (used by compat wrappers)."""
    if _OUTPUT_AS == "torch":
        return t
    if _OUTPUT_AS == "array":
        return t.cpu().numpy()
    if callable(_OUTPUT_AS):
        return _OUTPUT_AS(t)
    return t


def interruptible(fn, *args, **kwargs):
    """pylibraft.common.interruptible parity: run fn so that Ctrl-C cancels
    in-flight stream waits cooperatively (core.interruptible token)."""
    from raft_amd.core.interruptible import Interruptible

    tok = Interruptible()
    try:
        return fn(*args, **kwargs)
    except KeyboardInterrupt:
        tok.cancel()
        raise
