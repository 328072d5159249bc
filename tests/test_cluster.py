import pytest
import torch

from raft_amd.cluster import KMeans, KMeansParams, kmeans_fit, kmeans_predict
from raft_amd.neighbors.fused_l2nn import fused_l2nn
from raft_amd.random import make_blobs, RngState


class TestFusedL2NN:
    def test_matches_cdist(self):
        torch.manual_seed(0)
        x, y = torch.randn(40, 7), torch.randn(11, 7)
        dmin, amin = fused_l2nn(x, y)
        d = torch.cdist(x.double(), y.double()) ** 2
        torch.testing.assert_close(dmin.double(), d.min(dim=1).values, rtol=1e-5, atol=1e-5)
        assert torch.equal(amin, d.argmin(dim=1))


class TestKMeans:
    def test_recovers_blobs_kmeanspp(self):
        x, labels, centers = make_blobs(1200, 6, n_clusters=5, cluster_std=0.3,
                                        center_box=(-15, 15), state=RngState(seed=3))
        model = kmeans_fit(x, KMeansParams(n_clusters=5, max_iter=50, seed=1,
                                           init="kmeans++"))
        # every true center has a fitted centroid nearby
        d = torch.cdist(centers, model.centroids)
        assert d.min(dim=1).values.max() < 1.0
        assert model.inertia < 1200 * 6 * 0.3 ** 2 * 3

    def test_random_init_converges(self):
        x, labels, centers = make_blobs(1200, 6, n_clusters=5, cluster_std=0.3,
                                        center_box=(-15, 15), state=RngState(seed=3))
        model = kmeans_fit(x, KMeansParams(n_clusters=5, max_iter=50, seed=1,
                                           init="random"))
        # random init may hit a local optimum; inertia must still be far below
        # the trivial one-cluster solution
        x1 = kmeans_fit(x, KMeansParams(n_clusters=1, max_iter=2, seed=0, init="random"))
        assert model.inertia < 0.25 * x1.inertia

    def test_predict_consistent(self):
        x, _, _ = make_blobs(300, 4, n_clusters=3, state=RngState(seed=0))
        km = KMeans(n_clusters=3, max_iter=20, seed=0).fit(x)
        pred = km.predict(x)
        d = torch.cdist(x, km.cluster_centers_)
        assert torch.equal(pred, d.argmin(dim=1))

    def test_transform_shape(self):
        x, _, _ = make_blobs(100, 4, n_clusters=3, state=RngState(seed=0))
        km = KMeans(n_clusters=3, max_iter=5, seed=0).fit(x)
        t = km.transform(x)
        assert t.shape == (100, 3)

    def test_n_init_picks_best_restart(self):
        """n_init (reference kmeans_types n_init): best-of-N restarts."""
        x, _, _ = make_blobs(800, 8, n_clusters=6, cluster_std=0.4,
                             state=RngState(seed=7))
        m1 = kmeans_fit(x, KMeansParams(n_clusters=6, max_iter=15, seed=2))
        m4 = kmeans_fit(x, KMeansParams(n_clusters=6, max_iter=15, seed=2,
                                        n_init=4))
        assert m4.inertia <= m1.inertia + 1e-3

    def test_monotone_inertia(self):
        """EM iterations must not increase inertia."""
        x, _, _ = make_blobs(500, 5, n_clusters=4, state=RngState(seed=2))
        prev = None
        for iters in (1, 3, 8, 15):
            m = kmeans_fit(x, KMeansParams(n_clusters=4, max_iter=iters, seed=5,
                                           init="random", tol=0.0))
            if prev is not None:
                assert m.inertia <= prev * 1.001
            prev = m.inertia


class TestKMeansBalanced:
    def test_balanced_fit(self):
        from raft_amd.cluster import kmeans_balanced_fit
        x, _, centers = make_blobs(2000, 5, n_clusters=6, cluster_std=0.3,
                                   center_box=(-12, 12), state=RngState(seed=4))
        model = kmeans_balanced_fit(x, 6, max_iter=30, seed=1, sample_fraction=0.5)
        counts = torch.bincount(model.labels, minlength=6)
        assert (counts > 0).all()
        assert model.inertia < 2000 * 5 * 0.3 ** 2 * 5


class TestWeightedKMeans:
    def test_sample_weights_shift_centroids(self):
        # two tight blobs; heavily weight one point far away in cluster 0
        x = torch.cat([torch.randn(100, 2) * 0.05,
                       torch.randn(100, 2) * 0.05 + 10.0])
        x = torch.cat([x, torch.tensor([[4.0, 0.0]])])
        w = torch.ones(201)
        w[200] = 100.0
        init = torch.tensor([[0.0, 0.0], [10.0, 10.0]])
        m = kmeans_fit(x, KMeansParams(n_clusters=2, max_iter=10, init="array"),
                       init_centroids=init, sample_weights=w)
        # the heavy point drags centroid 0 toward (4, 0)
        assert m.centroids[0, 0] > 1.0
        m2 = kmeans_fit(x, KMeansParams(n_clusters=2, max_iter=10, init="array"),
                        init_centroids=init)
        assert m2.centroids[0, 0] < m.centroids[0, 0]


class TestKMeansScalable:
    def test_scalable_init_recovers_blobs(self):
        x, _, centers = make_blobs(5000, 8, n_clusters=10, cluster_std=0.3,
                                   center_box=(-15, 15), state=RngState(seed=4))
        m = kmeans_fit(x, KMeansParams(n_clusters=10, max_iter=30, seed=1,
                                       init="scalable", oversampling=2.0))
        d = torch.cdist(centers, m.centroids)
        assert d.min(dim=1).values.max() < 1.0
        assert m.inertia < 5000 * 8 * 0.3 ** 2 * 3

    def test_scalable_handles_tiny_k(self):
        x, _, _ = make_blobs(500, 4, n_clusters=2, state=RngState(seed=5))
        m = kmeans_fit(x, KMeansParams(n_clusters=2, max_iter=10, seed=0,
                                       init="kmeans||"))
        assert m.centroids.shape == (2, 4)


class TestEstimatorSurface:
    def test_fit_predict_transform_score(self):
        x, y, _ = make_blobs(600, 6, n_clusters=4, cluster_std=0.3,
                             state=RngState(seed=8))
        km = KMeans(n_clusters=4, max_iter=20, seed=1, n_init=2)
        labels = km.fit_predict(x)
        assert labels.shape == (600,)
        t = km.fit_transform(x)
        assert t.shape == (600, 4)
        s = km.score(x)
        assert s <= 0 and s == pytest.approx(-km.inertia_, rel=1e-3)
        assert km.labels_ is not None and km.n_iter_ >= 1


class TestHalfPrecisionAccumulation:
    def test_bf16_counts_and_centroids_exact(self):
        # bf16 accumulation made counts >256 inexact. One cluster
        # with 2000 members at a constant offset must produce the exact mean.
        n, d = 2000, 8
        x = torch.full((n, d), 1.0, dtype=torch.bfloat16)
        x[:, 0] = 3.0
        init = torch.zeros((1, d), dtype=torch.bfloat16)
        m = kmeans_fit(x, KMeansParams(n_clusters=1, max_iter=1, init="array"),
                       init_centroids=init)
        # mean of 2000 identical bf16 rows == the row itself, exactly
        assert torch.equal(m.centroids[0].float(), x[0].float())

    def test_bf16_iterate_matches_fp32(self):
        from raft_amd.cluster.kmeans import kmeans_iterate
        x32, _, centers = make_blobs(4000, 8, n_clusters=4, cluster_std=0.2,
                                     state=RngState(seed=11))
        c0 = centers.clone() + 0.05
        c32, i32 = kmeans_iterate(x32, c0.clone(), 2)
        c16, i16 = kmeans_iterate(x32.bfloat16(), c0.bfloat16(), 2)
        # quantization-level agreement only (inputs rounded to bf16), but the
        # old bf16 count accumulation was off by >10% on 1000-member clusters
        assert torch.allclose(c16.float(), c32, rtol=0.02, atol=0.05)
        assert abs(i16 - i32) / i32 < 0.05


class TestWeightedSeeding:
    def test_weighted_kmeanspp_prefers_heavy_points(self):
        # two tight blobs; blob A carries 100x the weight. With k=1 the single
        # center must land on A's side: the weighted potential demands it.
        torch.manual_seed(0)
        a = torch.randn(50, 2) * 0.05 + torch.tensor([5.0, 0.0])
        b = torch.randn(50, 2) * 0.05 + torch.tensor([-5.0, 0.0])
        x = torch.cat([a, b])
        w = torch.cat([torch.full((50,), 100.0), torch.full((50,), 1.0)])
        m = kmeans_fit(x, KMeansParams(n_clusters=1, max_iter=5, seed=3,
                                       init="kmeans++"), sample_weights=w)
        assert m.centroids[0, 0] > 4.0  # weighted mean ~ 4.9, unweighted ~ 0
