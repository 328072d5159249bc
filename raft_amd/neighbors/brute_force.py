"""Brute-force k-nearest-neighbors: pairwise distance tiles + select_k.

Reference parity: the historical raft::neighbors brute-force knn (tiled
distance + k-selection); BASELINE config 5 is this op at k=64 bf16 over 100M
rows chunked for 288 GB HBM.
"""
from __future__ import annotations

import torch

from raft_amd.distance import pairwise_distance, DistanceType
from raft_amd.matrix.select_k import select_k
from raft_amd.utils import row_chunks


def knn(x: torch.Tensor, queries: torch.Tensor, k: int,
        metric: DistanceType | str = DistanceType.L2Expanded,
        query_chunk: int = 16384, index_chunk: int = 262144,
        fp32_mode: str = "auto"):
    """k nearest rows of x for each query row. Returns (dists [q,k], idx [q,k]).

    Double-chunked (query rows x index rows) so the distance tile is bounded:
    tile bytes = query_chunk * index_chunk * 4 — sized for HBM3E residency.
    Per query chunk, partial top-k results from each index chunk are merged by
    a select_k over the concatenated candidates (k-way merge, the same scheme
    the reference uses for multi-block warpsort merges).
    """
    q = queries.shape[0]
    n = x.shape[0]
    # distances are fp32 on the GPU bf16 path (MFMA fp32 accumulate)
    out_dtype = torch.float32 if (queries.is_cuda and queries.dtype == torch.bfloat16) \
        else queries.dtype
    out_d = torch.empty((q, k), dtype=out_dtype, device=queries.device)
    out_i = torch.empty((q, k), dtype=torch.int64, device=queries.device)
    for qs, qe in row_chunks(q, query_chunk):
        cand_d, cand_i = [], []
        for is_, ie in row_chunks(n, index_chunk):
            dist = pairwise_distance(queries[qs:qe], x[is_:ie], metric=metric,
                                     fp32_mode=fp32_mode)
            kk = min(k, ie - is_)
            dd, ii = select_k(dist, kk, select_min=True)
            cand_d.append(dd)
            cand_i.append(ii + is_)
        dcat = torch.cat(cand_d, dim=1)
        icat = torch.cat(cand_i, dim=1)
        dd, pos = select_k(dcat, k, select_min=True)
        out_d[qs:qe] = dd
        out_i[qs:qe] = torch.gather(icat, 1, pos)
    return out_d, out_i
