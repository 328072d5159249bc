"""Linear assignment problem (LAP), single and batched.

Reference parity: raft/solver/linear_assignment.cuh (data-parallel BATCHED
Hungarian: 6-step machine over lap_kernels.cuh) driven from cuML/cuGraph.

MI355X design: the GPU-native LAP algorithm is the *auction* algorithm with
epsilon scaling — every unassigned row bids in parallel (one row-min/second-min
reduction per round, batched over the whole cost tile), which maps to wave64
reductions far better than Hungarian's serial augmenting paths. The batched
solver runs ALL problems' auction rounds in one tensor program (flat
(batch, col) scatter domain), and the convergence check runs once every 8
rounds instead of every round (the per-round `bool((assign <
0).any())` host sync serialized the pipeline). Exactness: for integer-scaled
costs, eps < 1/n yields the optimal assignment; we eps-scale down to that.
Validated against scipy's Hungarian in tests (single and batched).
"""
from __future__ import annotations

import torch


def _auction_phases(span: float, eps_final: float, eps_scale: float):
    phases = []
    e = span / 2.0
    while e > eps_final:
        phases.append(e)
        e *= eps_scale
    phases.append(eps_final)
    return phases


def _auction_batched(c: torch.Tensor, phases, check_every: int = 8,
                     max_rounds: int | None = None):
    """Run the eps-scaled auction on c [B, n, n] (MAXIMIZATION). Returns
    assign [B, n] (row -> col). All rounds are device-side; the host reads
    one boolean every `check_every` rounds."""
    B, n, _ = c.shape
    dev = c.device
    price = torch.zeros(B, n, dtype=torch.float64, device=dev)
    owner = torch.full((B, n), -1, dtype=torch.int64, device=dev)   # col -> row
    assign = torch.full((B, n), -1, dtype=torch.int64, device=dev)  # row -> col
    flat_rows = torch.arange(B * n, device=dev).reshape(B, n)
    limit = max_rounds or (n * n + 10 * n + 100)

    neg_inf = torch.full((), float("-inf"), dtype=torch.float64, device=dev)
    for e in phases:
        owner.fill_(-1)
        assign.fill_(-1)
        rounds = 0
        while rounds < limit:
            if not bool((assign < 0).any()):     # ONE host sync per chunk
                break
            # compact the unassigned set at the sync point; rows assigned
            # mid-chunk stay in the set with their bids masked inert
            nz = (assign < 0).nonzero(as_tuple=True)
            b_idx, r_idx = nz[0], nz[1]
            for _ in range(check_every):
                rounds += 1
                alive = assign[b_idx, r_idx] < 0
                value = c[b_idx, r_idx] - price[b_idx]          # [u, n]
                top2 = torch.topk(value, k=min(2, n), dim=1)
                best_j = top2.indices[:, 0]
                best_v = top2.values[:, 0]
                second_v = top2.values[:, 1] if n > 1 else best_v
                bids = price[b_idx, best_j] + (best_v - second_v) + e
                bids = torch.where(alive, bids, neg_inf)
                flat_col = b_idx * n + best_j                   # [u]
                # highest bid per (batch, column) wins
                bid_price = torch.full((B * n,), float("-inf"),
                                       dtype=torch.float64, device=dev)
                bid_price = bid_price.scatter_reduce(0, flat_col, bids,
                                                     reduce="amax")
                won = (bid_price > float("-inf")).nonzero(as_tuple=True)[0]
                is_winner = bids == bid_price[flat_col]
                # tie-break: lowest flat row id (== lowest row within batch)
                winner_row = torch.full((B * n,), B * n, dtype=torch.int64,
                                        device=dev)
                winner_row = winner_row.scatter_reduce(
                    0, flat_col[is_winner],
                    flat_rows[b_idx[is_winner], r_idx[is_winner]],
                    reduce="amin")
                # evict previous owners of won columns (flat views)
                owner_f = owner.view(-1)
                assign_f = assign.view(-1)
                price_f = price.view(-1)
                prev = owner_f[won]
                evicted = prev[prev >= 0]
                assign_f[evicted] = -1
                rows_w = winner_row[won]                        # flat row ids
                owner_f[won] = rows_w
                assign_f[rows_w] = won - (rows_w // n) * n      # col in batch
                price_f[won] = bid_price[won]
    return assign


def linear_assignment(cost: torch.Tensor, eps_scale: float = 0.15,
                      max_rounds: int | None = None, tol: float | None = None):
    """Minimize sum cost[i, assign[i]] over permutations.

    Returns (row_assignment [n] int64, total_cost float).

    Auction theory: the final assignment is within n*eps_final of optimal.
    The classic "eps < 1/n => exact" bound holds for INTEGER costs only;
    for float costs we eps-scale down to eps_final = tol/n where tol is the
    requested ABSOLUTE optimality gap (default span * 1e-9).
    """
    assert cost.dim() == 2 and cost.shape[0] == cost.shape[1], "square cost matrix"
    assign, total = linear_assignment_batched(cost.unsqueeze(0),
                                              eps_scale=eps_scale,
                                              max_rounds=max_rounds, tol=tol)
    return assign[0], float(total[0])


def linear_assignment_batched(costs: torch.Tensor, eps_scale: float = 0.15,
                              max_rounds: int | None = None,
                              tol: float | None = None):
    """Solve B independent n x n assignment problems in one device program
    (reference parity: LinearAssignmentProblem's batched solve).

    Returns (assign [B, n] int64, total_costs [B] float64 tensor).
    """
    assert costs.dim() == 3 and costs.shape[1] == costs.shape[2], \
        "costs must be [batch, n, n]"
    B, n, _ = costs.shape
    c = -costs.double()                  # auction maximizes value
    span = float((c.max() - c.min()).item()) or 1.0
    if tol is None:
        tol = span * 1e-9 + 1e-12
    eps_final = max(tol / n, 1e-14)
    phases = _auction_phases(span, eps_final, eps_scale)
    assign = _auction_batched(c, phases, max_rounds=max_rounds)
    rows = torch.arange(n, device=costs.device)
    total = costs.double()[torch.arange(B, device=costs.device).unsqueeze(1),
                           rows.unsqueeze(0), assign].sum(dim=1)
    return assign, total


class LinearAssignmentProblem:
    """Class wrapper mirroring the reference's LinearAssignmentProblem
    (linear_assignment.cuh:60), including batched solve."""

    def __init__(self, n: int):
        self.n = n
        self.assignment_: torch.Tensor | None = None
        self.obj_: float | None = None

    def solve(self, cost: torch.Tensor):
        if cost.dim() == 3:
            assert cost.shape[1:] == (self.n, self.n)
            self.assignment_, obj = linear_assignment_batched(cost)
            self.obj_ = obj
            return self.assignment_
        assert cost.shape == (self.n, self.n)
        self.assignment_, self.obj_ = linear_assignment(cost)
        return self.assignment_

    def get_primal_objective(self):
        return self.obj_
