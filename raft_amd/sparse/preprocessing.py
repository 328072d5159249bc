"""TF-IDF and BM25 encodings over CSR term-document matrices.

Reference parity: raft/sparse/matrix/preprocessing.cuh:28-94 (fit idf stats +
transform values). Convention: rows = documents, cols = terms, values = counts.
"""
from __future__ import annotations

import math

import torch

from .types import CSR


def _doc_lengths(a: CSR) -> torch.Tensor:
    seg = torch.repeat_interleave(torch.arange(a.n_rows, device=a.device),
                                  (a.indptr[1:] - a.indptr[:-1]).to(torch.int64))
    dl = torch.zeros(a.n_rows, dtype=a.values.dtype, device=a.device)
    dl.index_add_(0, seg, a.values)
    return dl, seg


def _idf(a: CSR) -> torch.Tensor:
    """idf(t) = log((N+1)/(df+1)) + 1 (smoothed, reference formula)."""
    df = torch.zeros(a.n_cols, dtype=torch.float64, device=a.device)
    df.index_add_(0, a.indices.to(torch.int64),
                  torch.ones(a.nnz, dtype=torch.float64, device=a.device))
    n = float(a.n_rows)
    return (torch.log((n + 1.0) / (df + 1.0)) + 1.0)


def tfidf_transform(a: CSR) -> CSR:
    """TF-IDF weighting of a term-document CSR (reference preprocessing)."""
    idf = _idf(a)
    vals = a.values.double() * idf[a.indices.to(torch.int64)]
    return CSR(a.indptr, a.indices, vals.to(a.values.dtype), a.n_rows, a.n_cols)


def bm25_transform(a: CSR, k1: float = 1.6, b: float = 0.75) -> CSR:
    """BM25 weighting of a term-document CSR (reference preprocessing)."""
    dl, seg = _doc_lengths(a)
    avgdl = dl.mean().clamp_min(1e-12)
    idf = _idf(a)
    tf = a.values.double()
    dl_ratio = (dl[seg] / avgdl).double()
    score = idf[a.indices.to(torch.int64)] * (tf * (k1 + 1.0)) / (
        tf + k1 * (1.0 - b + b * dl_ratio))
    return CSR(a.indptr, a.indices, score.to(a.values.dtype), a.n_rows, a.n_cols)
