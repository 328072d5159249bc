import numpy as np
import pytest
import torch

from raft_amd.distance import pairwise_distance, DistanceType


def _np_l2(x, y, squared=True):
    d = ((x[:, None, :] - y[None, :, :]) ** 2).sum(-1)
    return d if squared else np.sqrt(d)


class TestPairwise:
    def test_l2_vs_numpy(self):
        """BASELINE config 1: make_blobs-style data, CPU path, NumPy oracle."""
        torch.manual_seed(0)
        x = torch.randn(50, 13)
        y = torch.randn(30, 13)
        d = pairwise_distance(x, y, DistanceType.L2Expanded)
        ref = _np_l2(x.numpy().astype(np.float64), y.numpy().astype(np.float64))
        np.testing.assert_allclose(d.numpy(), ref, rtol=1e-4, atol=1e-4)

    def test_l2_sqrt(self):
        x = torch.randn(20, 5)
        d = pairwise_distance(x, x, DistanceType.L2SqrtExpanded)
        assert d.diagonal().abs().max() < 1e-3
        torch.testing.assert_close(d, d.t(), rtol=1e-4, atol=1e-4)

    def test_cosine(self):
        x = torch.randn(10, 6)
        d = pairwise_distance(x, x, DistanceType.CosineExpanded)
        assert d.diagonal().abs().max() < 1e-5
        assert (d >= -1e-6).all() and (d <= 2 + 1e-6).all()

    def test_inner_product(self):
        x, y = torch.randn(7, 4), torch.randn(9, 4)
        torch.testing.assert_close(pairwise_distance(x, y, DistanceType.InnerProduct),
                                   x @ y.t())

    @pytest.mark.parametrize("metric,ord_", [(DistanceType.L1, 1),
                                             (DistanceType.Linf, float("inf"))])
    def test_minkowski_family(self, metric, ord_):
        x, y = torch.randn(8, 5).double(), torch.randn(6, 5).double()
        d = pairwise_distance(x, y, metric)
        ref = torch.cdist(x, y, p=1 if ord_ == 1 else float("inf"))
        torch.testing.assert_close(d, ref, rtol=1e-6, atol=1e-8)

    def test_lp(self):
        x, y = torch.randn(5, 4).double(), torch.randn(5, 4).double()
        d = pairwise_distance(x, y, DistanceType.LpUnexpanded, p=3.0)
        ref = torch.cdist(x, y, p=3.0)
        torch.testing.assert_close(d, ref, rtol=1e-6, atol=1e-8)

    def test_hamming(self):
        x = torch.tensor([[1.0, 0, 1], [0, 0, 1]])
        d = pairwise_distance(x, x, DistanceType.HammingUnexpanded)
        torch.testing.assert_close(d, torch.tensor([[0.0, 1 / 3], [1 / 3, 0.0]]))

    def test_string_metric_names(self):
        x = torch.randn(4, 3)
        torch.testing.assert_close(pairwise_distance(x, x, "sqeuclidean"),
                                   pairwise_distance(x, x, DistanceType.L2Expanded))

    def test_bf16x3_mode_matches_native_fp32(self):
        torch.manual_seed(1)
        x, y = torch.randn(40, 32), torch.randn(24, 32)
        d_native = pairwise_distance(x, y, DistanceType.L2Expanded, fp32_mode="native")
        d_emul = pairwise_distance(x, y, DistanceType.L2Expanded, fp32_mode="bf16x3")
        ref = _np_l2(x.numpy().astype(np.float64), y.numpy().astype(np.float64))
        err_native = np.abs(d_native.numpy() - ref).max()
        err_emul = np.abs(d_emul.numpy() - ref).max()
        assert err_emul < max(err_native * 10, 2e-4)  # fp32-class accuracy
