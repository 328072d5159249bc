"""Minimum spanning tree / forest via Borůvka.

Reference parity: raft/sparse/solver/mst (mst_kernels.cuh: per-vertex min edge
:19, per-supervertex min :100, color propagation min_pair_colors/update_colors
:208-241 to fixpoint, weight alteration tie-break jitter :289; driver
mst_solver_inl.cuh).

Vectorized Borůvka over COO edges: each round (1) every component finds its
minimum outgoing edge (segmented amin via scatter_reduce), (2) those edges
join the MST, (3) components merge; label propagation runs pointer-jumping to
a fixpoint — the same min-propagation scheme as the reference's color kernels.
Tie-breaking uses a per-edge-id epsilon (alteration_kernel parity) which makes
the argmin unique and the result deterministic.
"""
from __future__ import annotations

import torch

from ..types import COO, CSR
from ..convert import csr_to_coo


def mst(a, symmetrize: bool = True):
    """MST/MSF of a weighted undirected graph (CSR or COO).

    Returns (src, dst, weight) of chosen edges (each undirected edge once).
    """
    if isinstance(a, CSR):
        a = csr_to_coo(a)
    n = max(a.n_rows, a.n_cols)
    rows = a.rows.to(torch.int64)
    cols = a.cols.to(torch.int64)
    w = a.values.double()
    orig_m = rows.numel()
    if symmetrize:
        rows, cols = torch.cat([rows, cols]), torch.cat([cols, rows])
        w = torch.cat([w, w])
    device = rows.device
    m = rows.numel()

    # deterministic tie-break jitter: unique (weight, edge_id) ranking
    eid = torch.arange(m, device=device, dtype=torch.float64)
    undirected_id = eid % orig_m if symmetrize else eid
    span = float((w.max() - w.min()).item()) if m else 0.0
    wj = w + undirected_id * (max(span, 1.0) * 1e-14)

    color = torch.arange(n, device=device)
    chosen = torch.zeros(m, dtype=torch.bool, device=device)
    inf = float("inf")

    for _ in range(64):  # <= log2(n) Borůvka rounds
        cr, cc = color[rows], color[cols]
        cross = cr != cc
        if not bool(cross.any()):
            break
        # (1) min outgoing edge weight per component
        minw = torch.full((n,), inf, dtype=torch.float64, device=device)
        minw = minw.scatter_reduce(0, cr[cross], wj[cross], reduce="amin")
        is_min = cross & (wj == minw[cr])          # unique thanks to jitter
        sel = is_min.nonzero(as_tuple=True)[0]
        chosen[sel] = True
        # (2) merge: each component's root points at its partner's root
        parent = torch.arange(n, device=device)
        parent[cr[sel]] = cc[sel]
        # break 2-cycles a<->b: root at the smaller color id
        two = parent[parent] == torch.arange(n, device=device)
        parent = torch.where(two & (parent > torch.arange(n, device=device)),
                             torch.arange(n, device=device), parent)
        # (3) pointer-jump to fixpoint
        for _ in range(64):
            nxt = parent[parent]
            if bool(torch.equal(nxt, parent)):
                break
            parent = nxt
        color = parent[color]

    sel = chosen.nonzero(as_tuple=True)[0]
    u = torch.minimum(rows[sel], cols[sel])
    v = torch.maximum(rows[sel], cols[sel])
    key = u * n + v
    _, first_idx = _unique_first(key)
    keep = sel[first_idx]
    return rows[keep], cols[keep], w[keep].to(a.values.dtype)


def _unique_first(key: torch.Tensor):
    """unique keys + index of first occurrence (vectorized)."""
    sorted_key, order = torch.sort(key, stable=True)
    is_first = torch.ones_like(sorted_key, dtype=torch.bool)
    is_first[1:] = sorted_key[1:] != sorted_key[:-1]
    return sorted_key[is_first], order[is_first]
