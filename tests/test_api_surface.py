"""Thin-wrapper API surface coverage: every public symbol gets at least one
behavioral check (guards against silent re-export breakage and signature
drift in the long tail of small reference-parity functions)."""
import pytest
import torch


class TestLinalgSurface:
    def test_map_family(self):
        from raft_amd.linalg import (map_op, unary_op, binary_op, ternary_op,
                                     subtract, power, eltwise, map_reduce)
        x = torch.tensor([1.0, 4.0])
        y = torch.tensor([2.0, 2.0])
        torch.testing.assert_close(map_op(lambda a, b: a + b, x, y), x + y)
        torch.testing.assert_close(unary_op(lambda a: a * 2, x), x * 2)
        torch.testing.assert_close(binary_op(lambda a, b: a * b, x, y), x * y)
        torch.testing.assert_close(ternary_op(lambda a, b, c: a + b + c, x, y, x),
                                   x + y + x)
        torch.testing.assert_close(subtract(x, y), x - y)
        torch.testing.assert_close(power(x, 2.0), x ** 2)
        torch.testing.assert_close(eltwise(lambda a: a + 1, x), x + 1)
        assert float(map_reduce(lambda a: a * a, "sum", x)) == pytest.approx(17.0)

    def test_misc_wrappers(self):
        from raft_amd.linalg import (col_norm, linewise_op, eig_jacobi,
                                     mean_squared_error, init_iota, init_eye,
                                     tsvd_fit, tsvd_transform)
        x = torch.randn(6, 4)
        torch.testing.assert_close(col_norm(x), x.norm(dim=0))
        v = torch.randn(4)
        torch.testing.assert_close(linewise_op(x, v, fn=lambda a, b: a + b), x + v)
        sym = x.T @ x
        w, q = eig_jacobi(sym)
        torch.testing.assert_close(q @ torch.diag(w) @ q.T, sym, atol=1e-4, rtol=1e-4)
        assert float(mean_squared_error(x, x)) == 0.0
        torch.testing.assert_close(init_iota(4, 1.0, 2.0),
                                   torch.tensor([1.0, 3.0, 5.0, 7.0]))
        assert torch.equal(init_eye(3), torch.eye(3))
        m = tsvd_fit(x, 2)
        assert tsvd_transform(m, x).shape == (6, 2)


class TestMatrixSurface:
    def test_ops(self):
        from raft_amd.matrix import (lower_triangular, power, matrix_sqrt,
                                     linewise, l2_norm)
        x = torch.rand(4, 4) + 0.1
        assert torch.equal(lower_triangular(x), torch.tril(x))
        torch.testing.assert_close(power(x, 3.0), x ** 3)
        torch.testing.assert_close(matrix_sqrt(x), x.sqrt())
        v = torch.randn(4)
        torch.testing.assert_close(linewise(x, v, "add"), x + v)
        torch.testing.assert_close(l2_norm(x), x.norm(dim=1))


class TestSparseSurface:
    def test_masked_ops(self):
        from raft_amd.core import Bitset
        from raft_amd.sparse import (masked_matmul, laplacian_normalized,
                                     knn_graph_symmetrize, coo_sort, csr_row_op,
                                     CSR, COO, coo_to_csr)
        a = torch.randn(4, 3)
        b = torch.randn(5, 3)
        mask = Bitset(20, default=False)
        mask.set(torch.tensor([0, 6, 12, 19]))  # (0,0),(1,1),(2,2),(3,4)
        out = masked_matmul(a, b, mask)
        full = a @ b.T
        dense = out.to_torch_sparse().to_dense()
        assert dense[0, 0] == pytest.approx(float(full[0, 0]), abs=1e-5)
        assert dense[0, 1] == 0.0
        # normalized laplacian of a path graph: diag == 1
        adj = torch.tensor([[0.0, 1, 0], [1, 0, 1], [0, 1, 0]])
        ln = laplacian_normalized(CSR.from_dense(adj))
        ld = ln.to_torch_sparse().to_dense()
        torch.testing.assert_close(torch.diagonal(ld), torch.ones(3))
        # knn graph symmetrize: result contains both (i,j) and (j,i)
        idx = torch.tensor([[1], [0], [0]])
        dist = torch.ones(3, 1)
        g = knn_graph_symmetrize(idx, dist)
        gd = g.to_dense()
        assert torch.equal(gd, gd.T)
        # coo_sort orders rows; csr_row_op applies per-row fn
        coo = COO(torch.tensor([1, 0]).int(), torch.tensor([0, 1]).int(),
                  torch.tensor([2.0, 3.0]), 2, 2)
        cs = coo_sort(coo)
        assert cs.rows.tolist() == [0, 1]
        csr = coo_to_csr(coo)
        doubled = csr_row_op(csr, lambda vals, seg: vals * 2)
        torch.testing.assert_close(doubled.values, csr.values * 2)


class TestCoreStatsRandomSurface:
    def test_core_bits(self):
        from raft_amd.core import (HipError, DeviceResourcesSNMG,
                                   interruptible_synchronize, get_logger, set_level)
        assert issubclass(HipError, RuntimeError)
        snmg = DeviceResourcesSNMG(device_ids=[])  # CPU container: empty set
        assert snmg is not None
        interruptible_synchronize()          # CPU: no-op, must not raise
        log = get_logger()
        set_level("warn")
        log.warning("surface check")

    def test_stats_random_bits(self):
        from raft_amd.stats import sum_cols, mean_add, mean_center
        from raft_amd.random import uniform_int
        x = torch.randn(10, 3)
        torch.testing.assert_close(sum_cols(x), x.sum(0), atol=1e-5, rtol=1e-5)
        mu = x.mean(0)
        torch.testing.assert_close(mean_add(mean_center(x), mu), x,
                                   atol=1e-5, rtol=1e-5)
        r = uniform_int((100,), 3, 7)
        assert int(r.min()) >= 3 and int(r.max()) < 7

    def test_neighbors_cluster_bits(self):
        from raft_amd.neighbors import fused_l2nn_argmin
        from raft_amd.cluster import kmeans_fit, kmeans_transform, KMeansParams
        x = torch.randn(50, 8)
        y = torch.randn(6, 8)
        a = fused_l2nn_argmin(x, y)
        assert torch.equal(a, (torch.cdist(x, y) ** 2).argmin(dim=1))
        m = kmeans_fit(x, KMeansParams(n_clusters=3, max_iter=5, seed=0))
        t = kmeans_transform(m, x)
        assert t.shape == (50, 3)


class TestRound2Surface:
    """Round-2 additions stay importable/callable on CPU."""

    def test_mr_surface(self):
        from raft_amd.core import (PoolMemoryResource, LimitingAdaptor,
                                   TrackingAdaptor, TorchMemoryResource,
                                   MemoryLimitExceeded, WorkspaceBuffer,
                                   AllocationStats)
        mr = TrackingAdaptor(PoolMemoryResource(1 << 12, device="cpu"))
        with mr.allocate(64) as b:
            assert b.tensor.numel() >= 64
        mr.assert_no_leaks()

    def test_linewise_fused_cpu(self):
        from raft_amd.linalg import linewise_fused
        x = torch.randn(8, 12)
        mu = torch.randn(12)
        sig = torch.rand(12) + 0.5
        out = linewise_fused(x, mu, "sub", sig, "div")
        torch.testing.assert_close(out, (x - mu) / sig)

    def test_philox_state(self):
        from raft_amd.random import RngState
        from raft_amd.random.rng import uniform
        u = uniform((16,), state=RngState(seed=1, gen_type="philox"))
        assert u.shape == (16,)

    def test_batched_lap_surface(self):
        from raft_amd.solver import linear_assignment_batched
        a, t = linear_assignment_batched(torch.rand(2, 6, 6))
        assert a.shape == (2, 6) and t.shape == (2,)

    def test_stats_round2(self):
        from raft_amd.stats import silhouette_score_batched, cov
        x = torch.randn(30, 4)
        c = cov(x, weights=torch.ones(30))
        assert c.shape == (4, 4)

    def test_comms_async_surface(self):
        from raft_amd.comms import LoopbackComms
        h = LoopbackComms().allreduce_async(torch.ones(3))
        assert h is None  # loopback: no handle

    def test_eig_variants_surface(self):
        from raft_amd.linalg.decomp import eig_jacobi, eig_selective
        a = torch.randn(6, 6, dtype=torch.float64)
        a = (a + a.t()) / 2
        w, v = eig_jacobi(a)
        assert w.shape == (6,)
        w2, _ = eig_selective(a, 2, largest=False)
        assert w2.shape == (2,)

    def test_adaptive_engine_surface(self):
        """Late-round-2 surfaces: the bf16x1v verified mode, its bound
        constants, and the kNN per-slice inflation table."""
        from raft_amd.neighbors.fused_l2nn import (_DEFAULT_BOUND, _MODE_BOUND,
                                                   _MODE_NSLICE, _VERIFY_MODES)
        assert _MODE_NSLICE["bf16x1v"] == 1 and "bf16x1v" in _VERIFY_MODES
        lead, tail = _MODE_BOUND["bf16x1v"]
        # the 1-slice bound is exactly 2^6 wider than the 2-slice bound
        assert lead == _DEFAULT_BOUND[0] * 64 and tail == _DEFAULT_BOUND[1] * 64
        from raft_amd.neighbors.brute_force import _KNN_BOUND, _slices_of
        assert set(_KNN_BOUND) == {1, 2, 3}
        assert _KNN_BOUND[1] == (lead, tail)
        x = torch.randn(8, 64)
        assert len(_slices_of(x, "auto")) == 1          # 1-slice auto filter
        assert len(_slices_of(x, "bf16x2")) == 2
        assert len(_slices_of(x, "bf16x3")) == 3
        back = sum(s.float() for s in _slices_of(x, "bf16x3"))
        assert torch.allclose(back, x, atol=1e-5)

    def test_gemm_guard_shape_logic(self):
        """linalg.gemm row-chunks >2^30-element outputs (vendor 32-bit
        C-index overflow guard) — CPU tensors bypass the guard and small
        GPU-shaped calls are untouched; verify the small-shape path here."""
        from raft_amd.linalg import gemm
        a = torch.randn(8, 16)
        b = torch.randn(16, 4)
        torch.testing.assert_close(gemm(a, b), a @ b)

    def test_cpp_resources_header_exists(self):
        import os
        p = os.path.join(os.path.dirname(os.path.dirname(
            os.path.abspath(__file__))), "include", "raft_amd", "core",
            "resources.hpp")
        src = open(p).read()
        assert "class device_resources" in src
        assert "get_stream_from_pool" in src

    def test_brute_force_index_surface(self):
        """build/search split (reference brute_force::build parity) — CPU
        falls back to the tiled path through knn()."""
        from raft_amd.neighbors import BruteForceIndex, brute_force_build, knn
        x = torch.randn(500, 32)
        q = torch.randn(20, 32)
        idx = brute_force_build(x)
        assert idx.n_rows == 500 and idx.dim == 32
        d, i = idx.search(q, 4)
        dr, ir = knn(x, q, 4)
        assert torch.equal(i, ir)
        torch.testing.assert_close(d, dr)
