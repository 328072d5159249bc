import numpy as np
import pytest
import torch
from scipy.optimize import linear_sum_assignment

from raft_amd.solver import linear_assignment, LinearAssignmentProblem
from raft_amd.label import make_monotonic, get_unique_labels, get_ovr_labels, merge_labels
from raft_amd.spectral import partition, analyze_partition, analyze_modularity
from raft_amd.sparse import CSR


class TestLAP:
    @pytest.mark.parametrize("n,seed", [(5, 0), (20, 1), (50, 2)])
    def test_optimal_vs_scipy(self, n, seed):
        rng = np.random.RandomState(seed)
        cost = rng.randint(0, 100, size=(n, n)).astype(np.float64)
        assign, total = linear_assignment(torch.from_numpy(cost))
        r, c = linear_sum_assignment(cost)
        ref = cost[r, c].sum()
        # auction with integer costs and eps < 1/n is exact
        assert total == pytest.approx(ref, abs=1e-6)
        # assignment is a permutation
        assert np.array_equal(np.sort(assign.numpy()), np.arange(n))

    def test_class_wrapper(self):
        cost = torch.tensor([[4.0, 1.0, 3.0], [2.0, 0.0, 5.0], [3.0, 2.0, 2.0]])
        lap = LinearAssignmentProblem(3)
        a = lap.solve(cost)
        r, c = linear_sum_assignment(cost.numpy())
        assert lap.get_primal_objective() == pytest.approx(cost.numpy()[r, c].sum())


class TestLabel:
    def test_make_monotonic(self):
        l = torch.tensor([10, 10, 42, 7, 42])
        m = make_monotonic(l)
        assert m.max() == 2
        assert torch.equal(m, torch.tensor([1, 1, 2, 0, 2]))

    def test_unique_and_ovr(self):
        l = torch.tensor([3, 1, 3, 2])
        assert torch.equal(get_unique_labels(l), torch.tensor([1, 2, 3]))
        ovr = get_ovr_labels(l, 3)
        assert torch.equal(ovr, torch.tensor([1, -1, 1, -1]))

    def test_merge_labels(self):
        # a: {0,1}{2,3}; b: {1,2}{0}{3} -> all connected through shared rows
        a = torch.tensor([0, 0, 1, 1])
        b = torch.tensor([0, 1, 1, 2])
        merged = merge_labels(a, b)
        assert merged.unique().numel() == 1
        # disjoint case
        a2 = torch.tensor([0, 0, 1, 1])
        b2 = torch.tensor([5, 5, 9, 9])
        m2 = merge_labels(a2, b2)
        assert m2.unique().numel() == 2
        assert m2[0] == m2[1] and m2[2] == m2[3] and m2[0] != m2[2]


def _two_cliques_graph():
    """Two 5-cliques joined by one weak edge."""
    n = 10
    dense = torch.zeros(n, n)
    for block in (range(0, 5), range(5, 10)):
        for i in block:
            for j in block:
                if i != j:
                    dense[i, j] = 1.0
    dense[4, 5] = dense[5, 4] = 0.01
    return CSR.from_dense(dense)


class TestSpectral:
    def test_partition_two_cliques(self):
        g = _two_cliques_graph()
        labels, w, v = partition(g, n_clusters=2, seed=0)
        l = labels.tolist()
        assert len(set(l[:5])) == 1 and len(set(l[5:])) == 1 and l[0] != l[9]

    def test_analyze_partition(self):
        g = _two_cliques_graph()
        labels = torch.tensor([0] * 5 + [1] * 5)
        cut, sizes = analyze_partition(g, labels)
        assert cut == pytest.approx(0.01, abs=1e-6)
        assert sizes == [5, 5]
        bad = torch.tensor([0, 1] * 5)
        cut_bad, _ = analyze_partition(g, bad)
        assert cut_bad > cut

    def test_modularity(self):
        g = _two_cliques_graph()
        good = torch.tensor([0] * 5 + [1] * 5)
        bad = torch.tensor([0, 1] * 5)
        assert analyze_modularity(g, good) > analyze_modularity(g, bad)


class TestLAPBatched:
    def test_batched_matches_scipy(self):
        from scipy.optimize import linear_sum_assignment
        from raft_amd.solver.lap import linear_assignment_batched
        torch.manual_seed(0)
        costs = torch.rand(6, 40, 40) * 10
        assign, totals = linear_assignment_batched(costs)
        for b in range(6):
            r, c = linear_sum_assignment(costs[b].numpy())
            ref = float(costs[b].numpy()[r, c].sum())
            assert abs(float(totals[b]) - ref) < 1e-4 * max(1.0, abs(ref))
            assert sorted(assign[b].tolist()) == list(range(40))

    def test_batched_class_wrapper(self):
        from raft_amd.solver.lap import LinearAssignmentProblem
        torch.manual_seed(1)
        costs = torch.rand(3, 16, 16)
        lap = LinearAssignmentProblem(16)
        a = lap.solve(costs)
        assert a.shape == (3, 16)
        assert lap.get_primal_objective().shape == (3,)

    def test_integer_costs_exact(self):
        from scipy.optimize import linear_sum_assignment
        from raft_amd.solver.lap import linear_assignment
        torch.manual_seed(2)
        ci = torch.randint(0, 5, (30, 30)).double()
        a, t = linear_assignment(ci)
        r, c = linear_sum_assignment(ci.numpy())
        assert abs(t - ci.numpy()[r, c].sum()) < 1e-9
