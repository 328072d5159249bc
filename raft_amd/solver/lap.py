"""Linear assignment problem (LAP).

Reference parity: raft/solver/linear_assignment.cuh (data-parallel batched
Hungarian: 6-step machine over lap_kernels.cuh) driven from cuML/cuGraph.

MI355X design: the GPU-native LAP algorithm is the *auction* algorithm with
epsilon scaling — every unassigned row bids in parallel (one row-min/second-min
reduction per round, batched over the whole cost tile), which maps to wave64
reductions far better than Hungarian's serial augmenting paths. Exactness: for
integer-scaled costs, eps < 1/n yields the optimal assignment; we run scaled
phases down to that. Validated against scipy's Hungarian in tests.
"""
from __future__ import annotations

import torch


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
    n = cost.shape[0]
    c = -cost.double()                   # auction maximizes value
    span = float((c.max() - c.min()).item()) or 1.0
    if tol is None:
        tol = span * 1e-9 + 1e-12
    eps_final = max(tol / n, 1e-14)

    price = torch.zeros(n, dtype=torch.float64, device=cost.device)
    owner = torch.full((n,), -1, dtype=torch.int64, device=cost.device)      # col -> row
    assign = torch.full((n,), -1, dtype=torch.int64, device=cost.device)     # row -> col

    phases = []
    e = span / 2.0
    while e > eps_final:
        phases.append(e)
        e *= eps_scale
    phases.append(eps_final)

    for e in phases:
        owner.fill_(-1)
        assign.fill_(-1)
        rounds = 0
        limit = max_rounds or (n * n + 10 * n + 100)
        while bool((assign < 0).any()) and rounds < limit:
            rounds += 1
            unassigned = (assign < 0).nonzero(as_tuple=True)[0]
            value = c[unassigned] - price.unsqueeze(0)          # [u, n]
            top2 = torch.topk(value, k=min(2, n), dim=1)
            best_j = top2.indices[:, 0]
            best_v = top2.values[:, 0]
            second_v = top2.values[:, 1] if n > 1 else best_v
            bids = price[best_j] + (best_v - second_v) + e
            # highest bid per column wins (scatter amax + match)
            bid_price = torch.zeros(n, dtype=torch.float64, device=cost.device)
            bid_price.fill_(float("-inf"))
            bid_price = bid_price.scatter_reduce(0, best_j, bids, reduce="amax")
            won_cols = (bid_price > float("-inf")).nonzero(as_tuple=True)[0]
            # winning row per column: pick the (unique-ized) first matching bidder
            is_winner = bids == bid_price[best_j]
            # tie-break: lowest row index wins
            winner_row = torch.full((n,), n, dtype=torch.int64, device=cost.device)
            winner_row = winner_row.scatter_reduce(0, best_j[is_winner],
                                                   unassigned[is_winner], reduce="amin")
            # evict previous owners of won columns
            prev = owner[won_cols]
            evicted = prev[prev >= 0]
            assign[evicted] = -1
            rows_w = winner_row[won_cols]
            owner[won_cols] = rows_w
            assign[rows_w] = won_cols
            price[won_cols] = bid_price[won_cols]
    total = float(cost.double()[torch.arange(n, device=cost.device), assign].sum())
    return assign, total


class LinearAssignmentProblem:
    """Class wrapper mirroring the reference's LinearAssignmentProblem."""

    def __init__(self, n: int):
        self.n = n
        self.assignment_: torch.Tensor | None = None
        self.obj_: float | None = None

    def solve(self, cost: torch.Tensor):
        assert cost.shape == (self.n, self.n)
        self.assignment_, self.obj_ = linear_assignment(cost)
        return self.assignment_

    def get_primal_objective(self) -> float:
        return self.obj_
