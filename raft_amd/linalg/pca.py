"""PCA and truncated SVD.

Reference parity: raft/linalg/detail/pca.cuh:113-298 (fit via covariance->eig
or svd paths, transform/inverse, explained variance) and detail/tsvd.cuh.
"""
from __future__ import annotations

from dataclasses import dataclass, field

import torch

from .decomp import svd_flip


@dataclass
class PCAModel:
    components: torch.Tensor          # (k, d) rows = principal axes
    explained_variance: torch.Tensor  # (k,)
    explained_variance_ratio: torch.Tensor
    singular_values: torch.Tensor
    mean: torch.Tensor                # (d,)
    noise_variance: float = 0.0


def pca_fit(x: torch.Tensor, n_components: int, algo: str = "eig") -> PCAModel:
    """Fit PCA (cov->eig or svd path; explained variance retained)."""
    n, d = x.shape
    mean = x.mean(dim=0)
    xc = x - mean
    if algo == "eig":
        # covariance -> eigendecomposition (reference default path)
        cov = (xc.t() @ xc) / (n - 1)
        w, v = torch.linalg.eigh(cov)              # ascending
        w = torch.flip(w, dims=[0])[:n_components].clamp_min(0)
        v = torch.flip(v, dims=[1])[:, :n_components]
        v, _ = svd_flip(v, v)
        components = v.t()
        singular = torch.sqrt(w * (n - 1))
    elif algo == "svd":
        u, s, vh = torch.linalg.svd(xc, full_matrices=False)
        u, vt = svd_flip(u[:, :n_components], vh.t()[:, :n_components])
        components = vt.t()
        singular = s[:n_components]
        w = (s[:n_components] ** 2) / (n - 1)
    else:
        raise ValueError(algo)
    total_var = xc.var(dim=0, unbiased=True).sum()
    ratio = w / total_var
    noise = float((total_var - w.sum()).clamp_min(0) / max(d - n_components, 1))
    return PCAModel(components, w, ratio, singular, mean, noise)


def pca_transform(model: PCAModel, x: torch.Tensor) -> torch.Tensor:
    """Project rows onto the fitted principal components."""
    return (x - model.mean) @ model.components.t()


def pca_inverse_transform(model: PCAModel, z: torch.Tensor) -> torch.Tensor:
    """Map scores back to the original feature space."""
    return z @ model.components + model.mean


@dataclass
class TSVDModel:
    components: torch.Tensor
    singular_values: torch.Tensor


def tsvd_fit(x: torch.Tensor, n_components: int) -> TSVDModel:
    """Truncated SVD: no centering (detail/tsvd.cuh)."""
    u, s, vh = torch.linalg.svd(x, full_matrices=False)
    u, vt = svd_flip(u[:, :n_components], vh.t()[:, :n_components])
    return TSVDModel(vt.t(), s[:n_components])


def tsvd_transform(model: TSVDModel, x: torch.Tensor) -> torch.Tensor:
    """Project rows onto the truncated-SVD components."""
    return x @ model.components.t()
