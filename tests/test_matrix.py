import pytest
import torch

from raft_amd import matrix
from raft_amd.matrix import SelectAlgo


class TestSelectK:
    @pytest.mark.parametrize("batch,n,k", [(4, 100, 5), (1, 1000, 64), (16, 37, 37)])
    @pytest.mark.parametrize("select_min", [True, False])
    def test_against_sort(self, batch, n, k, select_min):
        torch.manual_seed(0)
        x = torch.randn(batch, n)
        vals, idx = matrix.select_k(x, k, select_min=select_min)
        ref = torch.sort(x, dim=1, descending=not select_min).values[:, :k]
        torch.testing.assert_close(vals, ref)
        # indices must fetch the returned values
        torch.testing.assert_close(torch.gather(x, 1, idx), vals)


class TestGatherScatter:
    def test_gather(self):
        x = torch.randn(10, 4)
        idx = torch.tensor([3, 3, 0, 9])
        torch.testing.assert_close(matrix.gather(x, idx), x[idx])

    def test_gather_if(self):
        x = torch.arange(12, dtype=torch.float32).reshape(6, 2)
        idx = torch.tensor([0, 1, 2])
        stencil = torch.tensor([1.0, 0.0, 1.0])
        out = matrix.gather_if(x, idx, stencil, lambda s: s > 0.5)
        assert torch.equal(out[0], x[0])
        assert torch.equal(out[1], torch.zeros(2))

    def test_scatter(self):
        x = torch.randn(5, 3)
        perm = torch.tensor([4, 3, 2, 1, 0])
        out = matrix.scatter(x, perm)
        torch.testing.assert_close(out[perm], x)


class TestArgMinMax:
    def test_argmin_argmax(self):
        x = torch.randn(7, 13)
        torch.testing.assert_close(matrix.argmin(x), x.argmin(dim=1))
        torch.testing.assert_close(matrix.argmax(x), x.argmax(dim=1))


class TestStructural:
    def test_slice_diag_tri(self):
        x = torch.randn(6, 6)
        torch.testing.assert_close(matrix.slice_matrix(x, 1, 2, 4, 5), x[1:4, 2:5])
        torch.testing.assert_close(matrix.get_diagonal(x), torch.diagonal(x))
        assert torch.equal(matrix.upper_triangular(x), torch.triu(x))

    def test_reverse_shift(self):
        x = torch.arange(12, dtype=torch.float32).reshape(4, 3)
        assert torch.equal(matrix.row_reverse(x), torch.flip(x, [0]))
        assert torch.equal(matrix.col_reverse(x), torch.flip(x, [1]))
        sh = matrix.shift_rows(x, 1, fill_value=-1)
        assert torch.equal(sh[0], torch.full((3,), -1.0))
        assert torch.equal(sh[1:], x[:3])

    def test_math_ops(self):
        x = torch.tensor([[1.0, -4.0], [0.0, 9.0]])
        torch.testing.assert_close(matrix.reciprocal(x, thres=0.5),
                                   torch.tensor([[1.0, -0.25], [0.0, 1 / 9]]))
        th = matrix.threshold(x, 0.5)
        assert th[0, 1] == 0.0 and th[1, 1] == 9.0
        sf = matrix.sign_flip(x.clone())
        assert (sf.abs().max(dim=0).values == sf.max(dim=0).values).all()


class TestSort:
    def test_col_wise_sort(self):
        k = torch.randn(5, 9)
        v = torch.arange(45).reshape(5, 9)
        sk, sv = matrix.col_wise_sort(k, v)
        torch.testing.assert_close(sk, torch.sort(k, dim=1).values)
        torch.testing.assert_close(torch.gather(k, 1, torch.sort(k, dim=1).indices), sk)


class TestSampleRows:
    def test_sample_rows(self):
        x = torch.randn(50, 3)
        out = matrix.sample_rows(x, 10)
        assert out.shape == (10, 3)
        # every sampled row exists in x
        d = (out.unsqueeze(1) - x.unsqueeze(0)).abs().sum(-1).min(dim=1).values
        assert (d < 1e-6).all()


class TestInplace:
    def test_gather_scatter_inplace(self):
        from raft_amd.matrix import gather_inplace, scatter_inplace
        x = torch.arange(12, dtype=torch.float32).reshape(4, 3)
        perm = torch.tensor([2, 0, 3, 1])
        g = gather_inplace(x.clone(), perm)
        assert torch.equal(g, torch.arange(12, dtype=torch.float32).reshape(4, 3)[perm])
        s = scatter_inplace(x.clone(), perm)
        assert torch.equal(s[perm], torch.arange(12, dtype=torch.float32).reshape(4, 3))
