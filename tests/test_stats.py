import math

import numpy as np
import pytest
import torch

from raft_amd import stats
from raft_amd.random import make_blobs, RngState


class TestMoments:
    def test_mean_var_std(self):
        x = torch.randn(200, 6).double()
        torch.testing.assert_close(stats.mean(x), x.mean(dim=0), rtol=1e-10, atol=1e-10)
        torch.testing.assert_close(stats.vars_(x), x.var(dim=0, unbiased=True),
                                   rtol=1e-8, atol=1e-10)
        torch.testing.assert_close(stats.stddev(x), x.std(dim=0), rtol=1e-8, atol=1e-10)

    def test_meanvar_minmax(self):
        x = torch.randn(100, 4)
        mu, var = stats.meanvar(x)
        torch.testing.assert_close(mu, x.mean(dim=0), rtol=1e-5, atol=1e-6)
        mn, mx = stats.minmax(x)
        torch.testing.assert_close(mn, x.min(dim=0).values)
        torch.testing.assert_close(mx, x.max(dim=0).values)

    def test_weighted_mean(self):
        x = torch.randn(10, 5).double()
        w = torch.rand(5).double()
        torch.testing.assert_close(stats.weighted_mean(x, w),
                                   (x * w).sum(dim=1) / w.sum(), rtol=1e-10, atol=1e-12)

    def test_mean_center(self):
        x = torch.randn(50, 3)
        c = stats.mean_center(x)
        assert c.mean(dim=0).abs().max() < 1e-5


class TestCov:
    def test_vs_torch(self):
        x = torch.randn(300, 5).double()
        torch.testing.assert_close(stats.cov(x), torch.cov(x.t()), rtol=1e-8, atol=1e-10)


class TestHistogram:
    def test_counts(self):
        x = torch.rand(10000, 2)
        h = stats.histogram(x, 10, lo=0.0, hi=1.0)
        assert h.sum(dim=0).tolist() == [10000, 10000]
        assert (h > 600).all()  # roughly uniform


class TestContingencyAndClusteringMetrics:
    def test_contingency(self):
        a = torch.tensor([0, 0, 1, 1, 2])
        b = torch.tensor([1, 1, 0, 1, 2])
        c = stats.contingency_matrix(a, b)
        assert c[0, 1] == 2 and c[1, 0] == 1 and c[2, 2] == 1

    def test_ari_perfect_and_random(self):
        a = torch.tensor([0, 0, 1, 1, 2, 2])
        assert stats.adjusted_rand_index(a, a) == pytest.approx(1.0)
        perm = torch.tensor([2, 2, 0, 0, 1, 1])  # same partition, renamed
        assert stats.adjusted_rand_index(a, perm) == pytest.approx(1.0)

    def test_ari_vs_sklearn_formula(self):
        # hand-checked small case
        a = torch.tensor([0, 0, 1, 1])
        b = torch.tensor([0, 1, 0, 1])
        assert stats.adjusted_rand_index(a, b) == pytest.approx(-0.5, abs=1e-9)

    def test_rand_index(self):
        a = torch.tensor([0, 0, 1, 1])
        b = torch.tensor([0, 0, 1, 2])
        # pairs: (01) same-same, (23) same-diff, rest diff-diff
        assert stats.rand_index(a, b) == pytest.approx(5 / 6)

    def test_entropy_and_mi(self):
        a = torch.tensor([0, 0, 1, 1])
        assert stats.entropy(a) == pytest.approx(math.log(2))
        assert stats.mutual_info_score(a, a) == pytest.approx(math.log(2))
        b = torch.tensor([0, 1, 0, 1])
        assert stats.mutual_info_score(a, b) == pytest.approx(0.0, abs=1e-9)

    def test_homogeneity_completeness_v(self):
        truth = torch.tensor([0, 0, 1, 1])
        pred = torch.tensor([0, 1, 2, 3])  # fully split: homogeneous, incomplete
        assert stats.homogeneity_score(truth, pred) == pytest.approx(1.0)
        assert stats.completeness_score(truth, pred) < 1.0
        assert 0.0 <= stats.v_measure(truth, pred) <= 1.0

    def test_kl(self):
        p = torch.tensor([0.5, 0.5])
        q = torch.tensor([0.9, 0.1])
        ref = 0.5 * math.log(0.5 / 0.9) + 0.5 * math.log(0.5 / 0.1)
        assert stats.kl_divergence(p, q) == pytest.approx(ref, rel=1e-6)

    def test_silhouette_separated_blobs(self):
        x, y, _ = make_blobs(400, 4, n_clusters=3, cluster_std=0.2,
                             center_box=(-20, 20), state=RngState(seed=1))
        s = stats.silhouette_score(x, y, 3)
        assert s > 0.8

    def test_dispersion_positive(self):
        x, y, _ = make_blobs(200, 3, n_clusters=4, state=RngState(seed=2))
        assert stats.dispersion(x, y, 4) > 0


class TestRegressionMetrics:
    def test_r2(self):
        y = torch.randn(100).double()
        assert stats.r2_score(y, y) == pytest.approx(1.0)
        assert stats.r2_score(y, torch.full_like(y, float(y.mean()))) == pytest.approx(0.0, abs=1e-9)

    def test_regression_metrics(self):
        y = torch.tensor([1.0, 2.0, 3.0])
        p = torch.tensor([1.0, 3.0, 5.0])
        mae, mse, medae = stats.regression_metrics(y, p)
        assert mae == pytest.approx(1.0)
        assert mse == pytest.approx(5 / 3)
        assert medae == pytest.approx(1.0)

    def test_information_criterion(self):
        aic = stats.information_criterion(-10.0, 3, 50, "aic")
        bic = stats.information_criterion(-10.0, 3, 50, "bic")
        assert aic == pytest.approx(26.0)
        assert bic == pytest.approx(20 + 3 * math.log(50))


class TestClassification:
    def test_accuracy(self):
        a = torch.tensor([1, 2, 3, 4])
        b = torch.tensor([1, 2, 0, 4])
        assert stats.accuracy_score(a, b) == pytest.approx(0.75)


class TestNeighborhood:
    def test_recall(self):
        t = torch.tensor([[0, 1], [2, 3]])
        f = torch.tensor([[1, 5], [2, 3]])
        assert stats.neighborhood_recall(f, t) == pytest.approx(0.75)

    def test_trustworthiness_identity(self):
        x = torch.randn(60, 5)
        s = stats.trustworthiness_score(x, x.clone(), n_neighbors=5)
        assert s == pytest.approx(1.0, abs=1e-6)


class TestSklearnCrossValidation:
    """Second independent oracle: scikit-learn (the CPU torch path already
    pins against hand-derived formulas; sklearn guards formula drift)."""

    def test_clustering_metrics_vs_sklearn(self):
        import sklearn.metrics as skm
        from raft_amd import stats
        torch.manual_seed(0)
        a = torch.randint(0, 6, (3000,))
        b = (a + (torch.rand(3000) < 0.3).long() * torch.randint(0, 6, (3000,))) % 6
        an, bn = a.numpy(), b.numpy()
        assert stats.adjusted_rand_index(a, b) == pytest.approx(
            skm.adjusted_rand_score(an, bn), abs=1e-6)
        assert stats.rand_index(a, b) == pytest.approx(
            skm.rand_score(an, bn), abs=1e-6)
        assert stats.mutual_info_score(a, b) == pytest.approx(
            skm.mutual_info_score(an, bn), abs=1e-6)
        assert stats.homogeneity_score(a, b) == pytest.approx(
            skm.homogeneity_score(an, bn), abs=1e-6)
        assert stats.completeness_score(a, b) == pytest.approx(
            skm.completeness_score(an, bn), abs=1e-6)
        assert stats.v_measure(a, b) == pytest.approx(
            skm.v_measure_score(an, bn), abs=1e-6)

    def test_silhouette_r2_vs_sklearn(self):
        import sklearn.metrics as skm
        from raft_amd import stats
        from raft_amd.random import make_blobs, RngState
        x, y, _ = make_blobs(1500, 8, n_clusters=4, cluster_std=1.0,
                             state=RngState(seed=3))
        assert stats.silhouette_score(x, y, 4) == pytest.approx(
            skm.silhouette_score(x.numpy(), y.numpy()), abs=1e-4)
        yt = torch.randn(500)
        yp = yt + 0.3 * torch.randn(500)
        assert stats.r2_score(yt, yp) == pytest.approx(
            skm.r2_score(yt.numpy(), yp.numpy()), abs=1e-5)

    def test_accuracy_trustworthiness_vs_sklearn(self):
        import sklearn.metrics as skm
        from sklearn.manifold import trustworthiness as sk_trust
        from raft_amd import stats
        torch.manual_seed(1)
        yt = torch.randint(0, 3, (1000,))
        yp = torch.where(torch.rand(1000) < 0.8, yt, (yt + 1) % 3)
        assert stats.accuracy_score(yt, yp) == pytest.approx(
            skm.accuracy_score(yt.numpy(), yp.numpy()), abs=1e-6)
        x = torch.randn(400, 16)
        emb = torch.randn(400, 2)
        assert stats.trustworthiness_score(x, emb, n_neighbors=7) == pytest.approx(
            sk_trust(x.numpy(), emb.numpy(), n_neighbors=7), abs=1e-4)

    def test_regression_metrics_vs_sklearn(self):
        import sklearn.metrics as skm
        from raft_amd.stats import regression_metrics
        torch.manual_seed(2)
        yt = torch.randn(999)  # odd n: torch.median = lower middle, sklearn
        yp = yt + torch.randn(999) * 0.5   # interpolates — odd length matches
        mae, mse, medae = regression_metrics(yt, yp)
        assert mae == pytest.approx(skm.mean_absolute_error(yt, yp), abs=1e-6)
        assert mse == pytest.approx(skm.mean_squared_error(yt, yp), abs=1e-6)
        assert medae == pytest.approx(skm.median_absolute_error(yt, yp), abs=1e-6)

    def test_entropy_vs_scipy(self):
        from scipy.stats import entropy as sp_entropy
        from raft_amd.stats import entropy
        import numpy as np
        torch.manual_seed(3)
        labels = torch.randint(0, 7, (2000,))
        counts = torch.bincount(labels).numpy()
        assert entropy(labels) == pytest.approx(
            float(sp_entropy(counts / counts.sum())), abs=1e-6)


class TestWeightedCovBatchedSilhouette:
    def test_weighted_cov_matches_repetition(self):
        # integer weights == row repetition
        from raft_amd.stats import cov
        torch.manual_seed(0)
        x = torch.randn(50, 4, dtype=torch.float64)
        w = torch.randint(1, 4, (50,)).double()
        rep = torch.repeat_interleave(x, w.long(), dim=0)
        torch.testing.assert_close(cov(x, weights=w), cov(rep),
                                   rtol=1e-10, atol=1e-10)
        torch.testing.assert_close(cov(x, sample=False, weights=w),
                                   cov(rep, sample=False),
                                   rtol=1e-10, atol=1e-10)

    def test_batched_silhouette_equals_unbatched(self):
        from raft_amd.stats import silhouette_score, silhouette_score_batched
        from raft_amd.random import make_blobs, RngState
        x, y, _ = make_blobs(600, 5, n_clusters=4, cluster_std=0.4,
                             state=RngState(seed=2))
        full = silhouette_score(x, y)
        batched = silhouette_score_batched(x, y, batch_size=100)
        assert abs(full - batched) < 1e-9
        # sklearn cross-check
        from sklearn.metrics import silhouette_score as sk
        assert abs(full - sk(x.numpy(), y.numpy())) < 1e-4
