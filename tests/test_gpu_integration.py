"""GPU integration tests: composed algorithms running end-to-end on device
(clustering metrics over native pairwise kernels, spectral partition over the
native SpMV + Lanczos, brute-force knn over MFMA distance + radix select)."""
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def dev():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from raft_amd._ext import require_ext
    require_ext()
    return torch.device("cuda")


class TestStatsGpu:
    def test_silhouette_on_blobs(self, dev):
        from raft_amd.random import make_blobs, RngState
        from raft_amd.stats import silhouette_score
        x, y, _ = make_blobs(20000, 64, n_clusters=8, cluster_std=0.3,
                             center_box=(-20, 20), state=RngState(seed=1), device=dev)
        s = silhouette_score(x, y, 8)
        assert s > 0.8

    def test_moments_and_cov(self, dev):
        from raft_amd import stats
        x = torch.randn(50000, 32, device=dev)
        torch.testing.assert_close(stats.mean(x).double(), x.double().mean(0),
                                   rtol=1e-4, atol=1e-4)
        torch.testing.assert_close(stats.cov(x).double(), torch.cov(x.double().t()),
                                   rtol=1e-3, atol=1e-3)

    def test_clustering_metrics(self, dev):
        from raft_amd import stats
        a = torch.randint(0, 5, (10000,), device=dev)
        assert stats.adjusted_rand_index(a, a) == pytest.approx(1.0)
        b = torch.randint(0, 5, (10000,), device=dev)
        assert abs(stats.adjusted_rand_index(a, b)) < 0.05


class TestSpectralGpu:
    def test_partition_two_blocks(self, dev):
        from raft_amd.sparse import CSR
        from raft_amd.spectral import partition, analyze_partition
        n = 400
        dense = torch.zeros(n, n, device=dev)
        for lo, hi in ((0, n // 2), (n // 2, n)):
            blk = torch.rand(hi - lo, hi - lo, device=dev) < 0.2
            dense[lo:hi, lo:hi] = blk.float()
        dense = ((dense + dense.t()) > 0).float()
        dense.fill_diagonal_(0)
        dense[n // 2 - 1, n // 2] = dense[n // 2, n // 2 - 1] = 0.01
        g = CSR.from_dense(dense)
        g.indptr = g.indptr.to(torch.int32)
        g.indices = g.indices.to(torch.int32)
        labels, w, v = partition(g, 2, seed=1)
        l = labels.cpu()
        first, second = l[: n // 2], l[n // 2:]
        # each half is (almost entirely) one cluster
        assert (first == first.mode().values).float().mean() > 0.95
        assert (second == second.mode().values).float().mean() > 0.95
        assert first.mode().values != second.mode().values


class TestSolverGpu:
    def test_lap_on_gpu(self, dev):
        from raft_amd.solver import linear_assignment
        from scipy.optimize import linear_sum_assignment
        torch.manual_seed(0)
        cost = torch.randint(0, 1000, (64, 64), device=dev).float()
        assign, total = linear_assignment(cost)
        r, c = linear_sum_assignment(cost.cpu().numpy())
        assert total == pytest.approx(cost.cpu().numpy()[r, c].sum(), abs=1e-4)


class TestRandomGpu:
    def test_rmat_and_permute(self, dev):
        from raft_amd.random import rmat, permute, RngState
        src, dst = rmat(10, 10, 50000, state=RngState(seed=2), device=dev)
        assert src.max() < 1024 and dst.max() < 1024
        p = permute(100000, state=RngState(seed=3), device=dev)
        assert torch.equal(torch.sort(p).values, torch.arange(100000, device=dev))

    def test_sample_and_mvg(self, dev):
        from raft_amd.random import sample_without_replacement, multi_variable_gaussian, RngState
        idx = sample_without_replacement(100000, 1000, state=RngState(seed=4), device=dev)
        assert idx.unique().numel() == 1000
        mean = torch.tensor([0.0, 1.0], device=dev)
        cov = torch.tensor([[1.0, 0.3], [0.3, 1.0]], device=dev)
        s = multi_variable_gaussian(mean, cov, 50000, state=RngState(seed=5))
        torch.testing.assert_close(torch.cov(s.t()), cov, rtol=0.1, atol=0.05)


class TestSparseGpuExtra:
    def test_spmm_sddmm_symmetrize(self, dev):
        from raft_amd import sparse as rsp
        torch.manual_seed(0)
        dense = torch.randn(300, 200, device=dev)
        dense[torch.rand_like(dense) > 0.1] = 0
        csr = rsp.CSR.from_dense(dense)
        b = torch.randn(200, 16, device=dev)
        torch.testing.assert_close(rsp.spmm(csr, b).double(), dense.double() @ b.double(),
                                   rtol=1e-4, atol=1e-4)
        a2 = torch.randn(300, 8, device=dev)
        b2 = torch.randn(200, 8, device=dev)
        out = rsp.sddmm(a2, b2, csr)
        full = a2.double() @ b2.t().double()
        coo = rsp.csr_to_coo(csr)
        torch.testing.assert_close(out.values.double(),
                                   full[coo.rows.long(), coo.cols.long()],
                                   rtol=1e-4, atol=1e-4)

    def test_randomized_svds_gpu(self, dev):
        from raft_amd import sparse as rsp
        from raft_amd.sparse.solver import randomized_svds
        torch.manual_seed(1)
        u0 = torch.rand(400, 5, device=dev, dtype=torch.float64)
        v0 = torch.rand(5, 300, device=dev, dtype=torch.float64)
        dense = u0 @ v0
        csr = rsp.CSR.from_dense(dense)
        u, s, v = randomized_svds(csr, k=5, n_iter=6, seed=0)
        approx = (u * s.unsqueeze(0)) @ v.t()
        torch.testing.assert_close(approx, dense, rtol=1e-5, atol=1e-5)


class TestKnnGpu:
    def test_knn_bf16_recall(self, dev):
        from raft_amd.neighbors import knn
        torch.manual_seed(2)
        x = torch.randn(50000, 128, device=dev).bfloat16()
        q = x[:256].clone()
        d, i = knn(x, q, k=10, index_chunk=20000)
        assert (i[:, 0] == torch.arange(256, device=dev)).float().mean() > 0.99
        # recall vs exact fp32 knn
        ref = torch.cdist(q.float(), x.float()) ** 2
        ri = torch.topk(ref, 10, dim=1, largest=False).indices
        hits = (i.unsqueeze(2) == ri.unsqueeze(1)).any(2).float().mean()
        assert hits > 0.95


class TestKMeansEndToEnd:
    def test_full_fit_kmeanspp_quality(self, dev):
        """end-to-end kmeans_fit (init + EM + convergence) at medium scale."""
        from raft_amd.cluster import kmeans_fit, KMeansParams
        from raft_amd.random import make_blobs, RngState
        x, _, centers = make_blobs(200000, 128, n_clusters=64, cluster_std=0.5,
                                   center_box=(-25, 25), state=RngState(seed=13),
                                   device=dev)
        m = kmeans_fit(x, KMeansParams(n_clusters=64, max_iter=25, seed=2,
                                       init="kmeans++", n_init=3))
        d = torch.cdist(centers, m.centroids)
        # kmeans++ is O(log k)-approx, not exact: allow one unlucky blob
        matched = (d.min(dim=1).values < 2.0).sum()
        assert int(matched) >= 63, int(matched)
        # inertia must still be near the per-point noise floor
        opt = 200000 * 128 * 0.5 ** 2
        assert m.inertia < 3 * opt
        # weighted fit API also runs on GPU
        w = torch.rand(200000, device=dev) + 0.5
        m2 = kmeans_fit(x, KMeansParams(n_clusters=64, max_iter=5, seed=2,
                                        init="random"), sample_weights=w)
        assert torch.isfinite(m2.centroids).all()


class TestDecompGpu:
    def test_eig_svd_qr_lstsq(self, dev):
        from raft_amd.linalg import eigh, svd, qr, lstsq, cholesky_r1_update, cholesky
        torch.manual_seed(0)
        a = torch.randn(128, 64, device=dev)
        sym = a.T @ a
        w, v = eigh(sym)
        torch.testing.assert_close(v @ torch.diag(w) @ v.T, sym, atol=1e-2, rtol=1e-3)
        u, s, v = svd(a)     # returns V (A = U S V^T)
        torch.testing.assert_close(u @ torch.diag(s) @ v.T, a, atol=1e-3, rtol=1e-3)
        q, r = qr(a)
        torch.testing.assert_close(q @ r, a, atol=1e-3, rtol=1e-3)
        x_true = torch.randn(64, device=dev)
        b = a @ x_true
        for algo in ("qr", "svd-qr", "svd-jacobi", "eig"):
            xh = lstsq(a, b, algo=algo)
            torch.testing.assert_close(xh, x_true, atol=1e-2, rtol=1e-2)
        l = cholesky(sym + 64 * torch.eye(64, device=dev))
        x = torch.randn(64, device=dev)
        l2 = cholesky_r1_update(l.clone(), x.clone())
        torch.testing.assert_close(l2 @ l2.T, l @ l.T + torch.outer(x, x),
                                   atol=1e-2, rtol=1e-3)

    def test_rsvd_pca(self, dev):
        from raft_amd.linalg import rsvd, pca_fit, pca_transform, pca_inverse_transform
        torch.manual_seed(1)
        base = torch.randn(2000, 8, device=dev) @ torch.randn(8, 64, device=dev)
        x = base + 0.01 * torch.randn(2000, 64, device=dev)
        u, s, vt = rsvd(x, k=8)
        assert float(s[7] / s[0]) > 1e-3  # captured the rank-8 signal
        m = pca_fit(x, 8)
        z = pca_transform(m, x)
        xr = pca_inverse_transform(m, z)
        assert float((xr - x).norm() / x.norm()) < 0.05


class TestLanczosGpu:
    def test_eigsh_vs_dense(self, dev):
        from raft_amd.sparse import CSR
        from raft_amd.sparse.solver import eigsh
        torch.manual_seed(2)
        n = 400
        dense = torch.randn(n, n, device=dev)
        dense = (dense + dense.T) / 2
        dense = dense * (torch.rand(n, n, device=dev) < 0.05)
        dense = (dense + dense.T) / 2
        dense += torch.diag(torch.rand(n, device=dev) * 0.1)
        a = CSR.from_dense(dense)
        w, v = eigsh(a, k=4, maxiter=200)
        ref = torch.linalg.eigvalsh(dense)[:4]
        torch.testing.assert_close(w.to(ref.dtype), ref, atol=1e-3, rtol=1e-3)

    def test_cpp_cycle_matches_per_step_path(self, dev):
        """plain-CSR input drives the one-call C++ extension cycle
        (ext.lanczos_cycle_); a LinearOperator wrapping the SAME matrix
        takes the per-step fused path — eigenvalues must agree."""
        from raft_amd.sparse import CSR
        from raft_amd.sparse.solver import eigsh
        from raft_amd.sparse.solver.linear_operator import LinearOperator
        from raft_amd.sparse.linalg import spmv
        torch.manual_seed(5)
        n = 600
        dense = torch.randn(n, n, device=dev)
        dense = (dense + dense.T) / 2
        dense = dense * (torch.rand(n, n, device=dev) < 0.08)
        dense = (dense + dense.T) / 2
        dense += torch.diag(torch.rand(n, device=dev))
        a = CSR.from_dense(dense)
        w_cycle, _ = eigsh(a, k=5, maxiter=150)
        op = LinearOperator((n, n), lambda x: spmv(a, x), device=a.device,
                            dtype=a.values.dtype)
        w_step, _ = eigsh(op, k=5, maxiter=150)
        torch.testing.assert_close(w_cycle, w_step, atol=1e-4, rtol=1e-4)


class TestMstGpu:
    def test_total_weight_vs_scipy(self, dev):
        import numpy as np
        import scipy.sparse as sp
        import scipy.sparse.csgraph as csgraph
        from raft_amd.sparse import COO
        from raft_amd.sparse.solver import mst
        torch.manual_seed(3)
        n = 300
        dense = torch.rand(n, n) * (torch.rand(n, n) < 0.1)
        dense = torch.maximum(dense, dense.T)   # symmetric weights
        dense.fill_diagonal_(0)
        # ensure connectivity via a ring
        for i in range(n):
            dense[i, (i + 1) % n] = dense[(i + 1) % n, i] = 0.5 + 0.001 * i
        nz = dense.nonzero(as_tuple=False)
        coo = COO(nz[:, 0].to(torch.int32).to(dev), nz[:, 1].to(torch.int32).to(dev),
                  dense[nz[:, 0], nz[:, 1]].to(dev), n, n)
        src, dst, w = mst(coo, symmetrize=False)
        ref = csgraph.minimum_spanning_tree(sp.csr_matrix(dense.numpy()))
        assert src.numel() == n - 1
        assert abs(float(w.sum()) - ref.sum()) < 1e-3 * max(1.0, ref.sum())


class TestPreprocessingLabelsGpu:
    def test_tfidf_bm25_match_cpu(self, dev):
        from raft_amd.sparse import CSR, tfidf_transform, bm25_transform
        torch.manual_seed(4)
        dense = (torch.rand(50, 30) * 5).int().float() * (torch.rand(50, 30) < 0.3)
        a_cpu = CSR.from_dense(dense)
        a_gpu = CSR.from_dense(dense.to(dev))
        for fn in (tfidf_transform, bm25_transform):
            torch.testing.assert_close(fn(a_gpu).values.cpu(), fn(a_cpu).values,
                                       atol=1e-5, rtol=1e-5)

    def test_merge_labels_gpu(self, dev):
        from raft_amd.label import merge_labels, make_monotonic
        a = torch.tensor([0, 0, 1, 1, 2, 2], device=dev)
        b = torch.tensor([0, 1, 1, 2, 5, 5], device=dev)
        merged = merge_labels(a, b)
        # chain 0~1~2 joined through b; last pair separate? b joins 1&2 groups;
        # classes: {0,1,2,3} all chained via b labels 1 and 2 -> single class
        assert merged[:4].unique().numel() == 1
        mono = make_monotonic(torch.tensor([5, 9, 5], device=dev))
        assert mono.tolist() == [0, 1, 0]


class TestBalancedTrustGpu:
    def test_kmeans_balanced_gpu(self, dev):
        from raft_amd.cluster import kmeans_balanced_fit
        from raft_amd.random import make_blobs, RngState
        x, _, _ = make_blobs(100000, 64, n_clusters=16, cluster_std=0.4,
                             state=RngState(seed=6), device=dev)
        model = kmeans_balanced_fit(x, 16, max_iter=10, seed=0)
        counts = torch.bincount(model.labels, minlength=16).float()
        # balanced: no cluster more than 3x the mean size
        assert float(counts.max()) < 3.0 * float(counts.mean())
        assert torch.isfinite(model.centroids).all()

    def test_trustworthiness_gpu(self, dev):
        from raft_amd.stats import trustworthiness_score as trustworthiness
        torch.manual_seed(7)
        x = torch.randn(2000, 32, device=dev)
        # identity embedding is perfectly trustworthy
        t = trustworthiness(x, x.clone(), n_neighbors=8)
        assert t == pytest.approx(1.0, abs=1e-6)
        # random embedding is not
        t2 = trustworthiness(x, torch.randn(2000, 2, device=dev), n_neighbors=8)
        assert t2 < 0.8


class TestScalableInitGpu:
    def test_kmeans_scalable_gpu(self, dev):
        from raft_amd.cluster import kmeans_fit, KMeansParams
        from raft_amd.random import make_blobs, RngState
        x, _, centers = make_blobs(200000, 64, n_clusters=32, cluster_std=0.4,
                                   center_box=(-20, 20), state=RngState(seed=9),
                                   device=dev)
        m = kmeans_fit(x, KMeansParams(n_clusters=32, max_iter=20, seed=3,
                                       init="scalable"))
        d = torch.cdist(centers, m.centroids)
        assert int((d.min(dim=1).values < 2.0).sum()) >= 31
        assert m.inertia < 3 * 200000 * 64 * 0.4 ** 2


class TestKMeansWideDim:
    def test_fast_iterate_d512(self, dev):
        """d=512 routes update_verify to the MAX_DREG=16 variant; the fast
        EM loop must match the native-fp32 engine's assignments per step."""
        from raft_amd.cluster import kmeans_iterate
        from raft_amd.neighbors.fused_l2nn import fused_l2nn
        from raft_amd.random import make_blobs, RngState
        x, _, _ = make_blobs(200000, 512, n_clusters=128, cluster_std=1.0,
                             center_box=(-8, 8), state=RngState(seed=17),
                             device=dev)
        torch.manual_seed(0)
        c0 = x[torch.randperm(200000, device=dev)[:128]].clone()
        cv, inertia_v = kmeans_iterate(x, c0.clone(), 2, fp32_mode="bf16x2v")
        # per-step exactness: every chosen centroid's TRUE distance is
        # fp32-indistinguishable from the optimum (near-ties may resolve
        # differently than the native engine's expanded-form rounding)
        _, av = fused_l2nn(x, c0, fp32_mode="bf16x2v")
        idx = torch.randperm(200000, device=dev)[:4000]
        ref = torch.cdist(x[idx].double(), c0.double()) ** 2
        opt = ref.min(dim=1).values
        chosen = ref[torch.arange(4000, device=dev), av[idx]]
        assert float((chosen - opt).max()) < 5e-2
        _, an = fused_l2nn(x, c0, fp32_mode="native")
        assert float((av == an).float().mean()) > 0.9999
        # the fast loop's result is a valid EM trajectory: inertia decreases
        _, inertia_v2 = kmeans_iterate(x, cv, 1, fp32_mode="bf16x2v")
        assert inertia_v2 <= inertia_v * 1.0001


class TestKnnFp32Exactness:
    """the fp32 filtered path re-ranks candidates by EXACT fp32
    distances with a per-row margin proof — results must match the exact
    expanded-fp32 top-k, and fp32_mode='native' must skip the MFMA filter."""

    @pytest.mark.parametrize("mode", ["auto", "bf16x1v", "bf16x2"])
    def test_fp32_filtered_rerank_exact(self, dev, mode):
        """auto/bf16x1v = 1-slice filter with the wider 2^-7 threshold
        inflation; bf16x2 = the tighter 2-slice filter — all must land the
        exact fp32 top-k via the re-rank + margin proof."""
        from raft_amd.neighbors import knn
        torch.manual_seed(0)
        x = torch.randn(60000, 128, device=dev)
        q = torch.randn(2000, 128, device=dev)
        d, i = knn(x, q, k=16, fp32_mode=mode)       # filtered+rerank
        xn = (x * x).sum(1)
        qn = (q * q).sum(1)
        ref = (qn.unsqueeze(1) + xn.unsqueeze(0) - 2.0 * (q @ x.t())).clamp_min(0)
        rd, ri = torch.topk(ref, 16, dim=1, largest=False)
        agree = (i == ri).float().mean()
        assert float(agree) > 0.9995, float(agree)   # near-ties may reorder
        torch.testing.assert_close(d, rd, rtol=1e-4, atol=1e-3)

    @pytest.mark.parametrize("dt", [torch.float32, torch.bfloat16])
    def test_brute_force_index_matches_oneshot(self, dev, dt):
        """BruteForceIndex (build/search split, cached slices+norms) must
        return exactly what the one-shot knn() filtered path returns."""
        from raft_amd.neighbors import brute_force_build, knn
        torch.manual_seed(2)
        x = torch.randn(80000, 128, device=dev).to(dt)
        q = torch.randn(1000, 128, device=dev).to(
            torch.float32 if dt == torch.float32 else dt)
        idx = brute_force_build(x)
        d1, i1 = idx.search(q, 16)
        d2, i2 = idx.search(q, 16)          # second search reuses the cache
        dr, ir = knn(x, q, 16)
        assert torch.equal(i1, ir) and torch.equal(i2, ir)
        torch.testing.assert_close(d1, dr)

    def test_knn_fp16_input(self, dev):
        """fp16 index/queries widen exactly to fp32 and ride the filtered
        path — results must match the fp32 run on the same data."""
        from raft_amd.neighbors import knn
        torch.manual_seed(4)
        x = torch.randn(40000, 128, device=dev).half()
        q = torch.randn(300, 128, device=dev).half()
        d, i = knn(x, q, k=8)
        dr, ir = knn(x.float(), q.float(), k=8)
        assert torch.equal(i, ir)
        torch.testing.assert_close(d, dr)

    def test_knn_any_d_padding(self, dev):
        """d=100 now stays on the filtered path (zero-padded K) for both
        one-shot knn and BruteForceIndex — must match the exact fp32 top-k."""
        from raft_amd.neighbors import brute_force_build, knn
        torch.manual_seed(3)
        x = torch.randn(60000, 100, device=dev)
        q = torch.randn(500, 100, device=dev)
        d, i = knn(x, q, k=8)
        idx = brute_force_build(x)
        assert idx.dim == 100
        d2, i2 = idx.search(q, 8)
        ref = ((q*q).sum(1, keepdim=True) + (x*x).sum(1).unsqueeze(0)
               - 2.0 * (q @ x.t())).clamp_min(0)
        rd, ri = torch.topk(ref, 8, dim=1, largest=False)
        assert (i == ri).float().mean() > 0.999
        assert torch.equal(i, i2)
        torch.testing.assert_close(d, rd, rtol=1e-4, atol=1e-3)

    def test_fp32_native_mode_honored(self, dev):
        from raft_amd.neighbors import knn
        torch.manual_seed(1)
        x = torch.randn(20000, 64, device=dev)
        q = torch.randn(500, 64, device=dev)
        d_nat, i_nat = knn(x, q, k=8, fp32_mode="native")
        d_auto, i_auto = knn(x, q, k=8)
        assert (i_nat == i_auto).float().mean() > 0.999
        torch.testing.assert_close(d_nat, d_auto, rtol=1e-3, atol=1e-3)
