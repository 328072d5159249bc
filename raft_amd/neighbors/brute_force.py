"""Brute-force k-nearest-neighbors.

Reference parity: the historical raft::neighbors brute-force knn (tiled
distance + k-selection); BASELINE config 5 is this op at k=64 bf16 over 100M
rows chunked for 288 GB HBM.

MI355X design — the GPU path never materializes the distance matrix:
  1. SAMPLE: distances to a random ~n/1024 subset of the index give each
     query row a threshold (the j-th smallest sampled distance, j chosen so
     the full-index count below it is >= k with high probability).
  2. FILTER: one MFMA sweep over the whole index emits only candidates with
     d2 <= threshold into per-row bounded buffers (csrc/pairwise_mfma.hip
     pairwise_l2_filter_kernel) — expected O(k) emissions per row, so the
     6.5 TB of distance-tile traffic of the naive tiled path disappears.
  3. SELECT: native radix select_k over the tiny candidate buffers.
  4. fp32 inputs: the selected candidates are RE-RANKED by exact-fp32
     distances (gather + einsum over ~2k candidates/row) and a per-row margin
     check proves the exact top-k is inside the re-ranked set; rows failing
     the margin, or whose buffer under/overflowed (probabilistically rare),
     fall back to the exact tiled path. The returned neighbors are therefore
     exact w.r.t. fp32 expanded distances for every row (the same guarantee
     and mechanism as the verified kmeans L2-NN engine, csrc/kmeans.hip).
     fp32_mode="native" skips the MFMA filter entirely and runs the tiled
     fp32 rocBLAS path.
"""
from __future__ import annotations

import torch

from raft_amd._ext import require_ext
from raft_amd.distance import pairwise_distance, DistanceType
from raft_amd.matrix.select_k import select_k
from raft_amd.utils import row_chunks


def knn(x: torch.Tensor, queries: torch.Tensor, k: int,
        metric: DistanceType | str = DistanceType.L2Expanded,
        query_chunk: int | None = None, index_chunk: int | None = None,
        fp32_mode: str = "auto", res=None):
    """k nearest rows of x for each query row. Returns (dists [q,k], idx [q,k]).

    query_chunk/index_chunk default to sizes derived from the Resources
    workspace budget (reference: workspace-resource-driven batching) — set a
    workspace limit via res.set_workspace_limit(nbytes) to bound scratch.
    """
    if x.dtype == torch.float16 and x.is_cuda:
        # fp16 -> fp32 is exact: ride the fp32 filtered path (reference
        # supports half-precision knn; the bf16 split of an exact widening
        # keeps the same provable-inflation guarantees)
        return knn(x.float(), queries.float(), k, metric, query_chunk,
                   index_chunk, fp32_mode, res)
    from raft_amd.core.resources import get_resources
    res = get_resources(res if res is not None else queries.device)
    if query_chunk is None or index_chunk is None:
        qc, ic = _tiles_from_budget(res.workspace_budget(), queries.shape[0],
                                    x.shape[0], x.shape[1], k)
        query_chunk = query_chunk or qc
        index_chunk = index_chunk or ic
    filterable = (queries.is_cuda
                  and metric in (DistanceType.L2Expanded, "sqeuclidean")
                  and x.shape[0] >= 8 * k
                  and x.dtype in (torch.bfloat16, torch.float32)
                  and not (x.dtype == torch.float32 and fp32_mode == "native"))
    if filterable and x.shape[1] % 64 != 0:
        # zero feature columns change no distance: pad to the MFMA K
        # granularity and keep the filtered path (round-1 d-cliff)
        dp = (-x.shape[1]) % 64
        x = torch.nn.functional.pad(x, (0, dp))
        queries = torch.nn.functional.pad(queries, (0, dp))
    if filterable:
        return _knn_gpu_filtered(x, queries, k, fp32_mode, res=res)
    return _knn_tiled(x, queries, k, metric, query_chunk, index_chunk, fp32_mode)


class BruteForceIndex:
    """Prebuilt brute-force index: build once, search many times.

    Reference parity: the build/search split of raft's historical
    brute-force knn (index object holding the dataset + norms). Caches the
    bf16 filter slice(s) and the row norms of the index matrix, so repeated
    searches skip the per-call index split — at 100M x 128 fp32 the split
    alone is a measurable fraction of a one-shot knn() call.
    """

    def __init__(self, x: torch.Tensor, fp32_mode: str = "auto", res=None):
        self.dim_orig = x.shape[1]
        self.fp32_mode = fp32_mode
        self._filterable = (
            x.is_cuda and x.dtype in (torch.bfloat16, torch.float32)
            and not (x.dtype == torch.float32 and fp32_mode == "native"))
        if self._filterable and x.shape[1] % 64 != 0:
            # pad the MFMA K granularity ONCE at build (zero columns change
            # no distance); search pads queries to match
            x = torch.nn.functional.pad(x, (0, (-x.shape[1]) % 64))
        self.x = x
        if self._filterable:
            self.slices = _slices_of(x, fp32_mode)
            self.xn = _norms(x)
        else:
            self.slices = None
            self.xn = None

    @property
    def n_rows(self) -> int:
        return self.x.shape[0]

    @property
    def dim(self) -> int:
        return self.dim_orig

    def search(self, queries: torch.Tensor, k: int, res=None):
        """k nearest index rows per query. Returns (dists [q,k], idx [q,k])."""
        if queries.shape[1] != self.x.shape[1]:
            queries = torch.nn.functional.pad(
                queries, (0, self.x.shape[1] - queries.shape[1]))
        if (self._filterable and queries.is_cuda
                and self.x.shape[0] >= 8 * k):
            from raft_amd.core.resources import get_resources
            r = get_resources(res if res is not None else queries.device)
            return _knn_gpu_filtered(self.x, queries, k, self.fp32_mode,
                                     res=r, pre=(self.slices, self.xn))
        return knn(self.x, queries, k, fp32_mode=self.fp32_mode, res=res)


def brute_force_build(x: torch.Tensor, fp32_mode: str = "auto",
                      res=None) -> BruteForceIndex:
    """reference brute_force::build parity."""
    return BruteForceIndex(x, fp32_mode=fp32_mode, res=res)


def _tiles_from_budget(budget_bytes: int, q: int, n: int, d: int, k: int):
    """Size the (query_chunk x index_chunk) distance tile from the workspace
    budget: the fp32 tile is the dominant scratch of the tiled path; keep it
    under half the budget, with floors that keep the chip busy."""
    q_chunk = max(min(q, 16384), 1)
    max_tile = max(budget_bytes // 2, 1 << 22)
    ic_floor = min(max(n, 1), max(4 * k, 1024))
    per_row = 4 * q_chunk + 8  # tile column + select candidates
    index_chunk = max(min(min(max(n, 1), 262144), max_tile // per_row), ic_floor)
    if index_chunk == ic_floor:
        # the floor won: shrink the query chunk instead so the tile still fits
        q_chunk = max(min(q_chunk, max_tile // (4 * ic_floor + 8)), 64)
    return int(q_chunk), int(index_chunk)


def _slices_of(t: torch.Tensor, fp32_mode: str):
    from raft_amd.neighbors.fused_l2nn import split_bf16_slices
    if t.dtype == torch.bfloat16:
        return [t.contiguous()]
    # auto: the 1-slice filter (1/3 MFMA work, half the slice stream) — the
    # wider 2^-7 inflation admits a provable candidate superset and the
    # exact re-rank + margin proof keeps results exact-fp32 either way
    # (measured 30M x 128 randn: 61.1k vs 27.5k q/s, idx agreement 0.9999
    # with dist diffs at fp32 rounding). Explicit "bf16x2"/"bf16x3" keep
    # the tighter filters for near-tie-heavy corpora.
    nsl = {"bf16x3": 3, "bf16x1v": 1, "auto": 1}.get(fp32_mode, 2)
    return split_bf16_slices(t, nsl)


#: provable |Δdot| <= lead*sqrt(qn*xn) + tail*(qn+xn) envelopes per slice
#: count (same family as the verified L2NN engine, csrc/kmeans.hip)
_KNN_BOUND = {1: (2.0 ** -7, 2.0 ** -12),
              2: (2.0 ** -13, 2.0 ** -18),
              3: (2.0 ** -21, 2.0 ** -24)}


def _norms(t: torch.Tensor) -> torch.Tensor:
    if t.is_cuda and t.dtype == torch.bfloat16:
        # native kernel: a torch float() chain would materialize a 2x-size
        # fp32 copy (51 GB for the 100M-row index — allocation-bound)
        return require_ext().rows_sqnorm_bf16(t.contiguous())
    if t.is_cuda and t.dtype == torch.float32:
        return require_ext().reduce_rows(t.contiguous(), 1)
    return t.float().pow(2).sum(dim=1).contiguous()


def _knn_gpu_filtered(x, queries, k, fp32_mode, index_chunk: int = 4_000_000,
                      res=None, pre=None):
    """Query-blocked driver: splits queries so the candidate buffers
    (cap x 8 B/row) + the exact-rerank gather fit the workspace budget, then
    runs the sample->filter->select pipeline per block. pre (optional):
    (index_slices, index_norms) from a BruteForceIndex — skips the per-call
    index split."""
    from raft_amd.core.resources import get_resources
    res = get_resources(res if res is not None else queries.device)
    m = queries.shape[0]
    cap = max(16384, 32 * k)
    per_row = cap * 8 + min(2 * k, cap) * x.shape[1] * 4 + 64
    budget = max(res.workspace_budget(), 1 << 22)
    q_block = max(256, min(m, int(budget // (2 * per_row)) or 1))
    if q_block >= m:
        return _knn_gpu_filtered_block(x, queries, k, fp32_mode, index_chunk,
                                       res, pre=pre)
    out_d = torch.empty((m, k), dtype=torch.float32, device=queries.device)
    out_i = torch.empty((m, k), dtype=torch.int64, device=queries.device)
    for s0, s1 in row_chunks(m, q_block):
        dv, iv = _knn_gpu_filtered_block(x, queries[s0:s1], k, fp32_mode,
                                         index_chunk, res, pre=pre)
        out_d[s0:s1] = dv
        out_i[s0:s1] = iv
    return out_d, out_i


def _knn_gpu_filtered_block(x, queries, k, fp32_mode, index_chunk, res,
                            pre=None):
    ext = require_ext()
    m, d = queries.shape
    n = x.shape[0]
    dev = queries.device

    q_slices = _slices_of(queries, fp32_mode)
    qn = _norms(queries)
    pre_slices = pre[0] if pre is not None else None
    xn_full = pre[1] if pre is not None else _norms(x)

    # ---- 1. sample -> per-row thresholds -----------------------------------
    s = max(min(65536, n), n // 1024)
    gen = torch.Generator(device="cpu").manual_seed(0x5eed)
    sample_idx = (torch.randint(0, n, (s,), generator=gen)
                  .to(dev))
    if pre_slices is not None:
        sample_slices = [sl[sample_idx].contiguous() for sl in pre_slices]
    else:
        xs_sample = x[sample_idx].contiguous()
        sample_slices = _slices_of(xs_sample, fp32_mode)
    sn = xn_full[sample_idx].contiguous()
    j = max(2, (4 * k * s) // max(n, 1))
    thr = torch.empty(m, dtype=torch.float32, device=dev)
    for s0, s1 in row_chunks(m, 8192):
        ds = ext.pairwise_l2_mfma([t[s0:s1] for t in q_slices], sample_slices,
                                  qn[s0:s1].contiguous(), sn)
        jv, _ = select_k(ds, min(j, s), select_min=True)
        thr[s0:s1] = jv[:, -1]
    del ds
    # inflate thresholds by 2x the provable split-emulation error bound so a
    # true neighbor can never be filtered out by split rounding (one bound
    # covers the sampled threshold reading low, one the candidate reading
    # high). bf16 input (1 slice) has no split error. Same bound family as
    # the verified L2NN engine (csrc/kmeans.hip l2nn verify).
    nslice = len(q_slices)
    # bf16 input is exact in its own dtype (single slice, no split error);
    # fp32 input is EMULATED at any slice count — including the 1-slice
    # bf16x1v mode, whose wider 2^-7 bound still inflates thresholds by far
    # less than the inter-candidate spacing on realistic data
    emulated = queries.dtype != torch.bfloat16
    if emulated:
        xm = float(xn_full.max())
        lead, tail = _KNN_BOUND[nslice]
        thr = thr + 4.0 * (lead * torch.sqrt(qn.clamp_min(0) * xm)
                           + tail * (qn + xm))

    # ---- 2. filtered emission over the full index --------------------------
    # candidate buffers come from the workspace resource (tracked/capped)
    cap = max(16384, 32 * k)
    ws_d = res.get_workspace((m, cap), torch.float32)
    ws_i = res.get_workspace((m, cap), torch.int32)
    cand_d = ws_d.view((m, cap), torch.float32)
    cand_d.fill_(float("inf"))
    cand_i = ws_i.view((m, cap), torch.int32)
    cand_i.fill_(-1)
    cnt = torch.zeros(m, dtype=torch.int32, device=dev)
    for c0, c1 in row_chunks(n, index_chunk):
        if pre_slices is not None:
            chunk_slices = [sl[c0:c1] for sl in pre_slices]
        else:
            chunk_slices = _slices_of(x[c0:c1], fp32_mode)
        ext.pairwise_l2_filter(q_slices, chunk_slices, qn,
                               xn_full[c0:c1].contiguous(), thr,
                               cand_d, cand_i, cnt, c0)

    # ---- 3. select over candidates -----------------------------------------
    bad = (cnt < k) | (cnt > cap)
    if emulated:
        # fp32 input: candidate distances are split-bf16 EMULATED — re-rank
        # the top 2k candidates by EXACT fp32 distances, then prove per row
        # that no candidate outside the re-ranked set can be a true neighbor:
        # exact_d >= emul_d - eps for every candidate, so if the exact k-th
        # selected <= (emulated (2k)-th) - eps, the exact top-k is inside the
        # 2k set. Rows failing the margin join the exact-fallback set.
        k2 = min(2 * k, cap)
        evals, pos2 = select_k(cand_d, k2, select_min=True)
        idx2 = torch.gather(cand_i, 1, pos2.to(torch.int64)).to(torch.int64)
        safe_idx2 = idx2.clamp_min(0)           # -1 slots carry +inf distances
        xg = x[safe_idx2.reshape(-1)].reshape(m, k2, d)
        dots = torch.einsum("md,mkd->mk", queries.float(), xg.float())
        exact = (qn.unsqueeze(1) + xn_full[safe_idx2] - 2.0 * dots).clamp_min_(0)
        exact = torch.where(idx2 < 0, torch.full_like(exact, float("inf")), exact)
        vals, rpos = torch.sort(exact, dim=1)
        vals = vals[:, :k]
        idx = torch.gather(idx2, 1, rpos[:, :k])
        eps = lead * torch.sqrt(qn.clamp_min(0) * xm) + tail * (qn + xm)
        # rows with cnt <= k2 re-ranked their ENTIRE candidate set (complete
        # by threshold inflation) — exact with no margin needed; rows with
        # more candidates need the emulated-(k2)th margin to clear eps
        need_margin = cnt.to(torch.int64) > k2
        margin_fail = need_margin & ~(vals[:, -1] <= evals[:, -1] - eps)
        bad = bad | margin_fail
    else:
        vals, pos = select_k(cand_d, k, select_min=True)
        idx = torch.gather(cand_i, 1, pos.to(torch.int64)).to(torch.int64)

    ws_d.free()
    ws_i.free()

    # ---- 4. exact fallback for margin-failed / under/overflowed rows --------
    n_bad = int(bad.sum().item())
    if n_bad:
        rows = bad.nonzero(as_tuple=True)[0]
        # the tiled fallback runs the GEMM layer, which has no 1-slice mode
        # (it exists only as a filter inflation choice) — fall back at auto
        fb_mode = "auto" if fp32_mode == "bf16x1v" else fp32_mode
        bd, bi = _knn_tiled(x, queries[rows], k, DistanceType.L2Expanded,
                            8192, index_chunk, fb_mode)
        vals[rows] = bd.to(vals.dtype)
        idx[rows] = bi
    return vals, idx


def _knn_tiled(x, queries, k, metric, query_chunk, index_chunk, fp32_mode):
    """Exact tiled path (distance tiles + per-tile select + k-way merge)."""
    q = queries.shape[0]
    n = x.shape[0]
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
