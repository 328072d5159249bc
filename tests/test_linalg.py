import numpy as np
import pytest
import torch

from raft_amd import linalg
from raft_amd.linalg import Apply, NormType
from raft_amd.linalg.decomp import eig_jacobi, eig_selective


class TestReduce:
    @pytest.mark.parametrize("shape", [(4, 7), (1, 1), (33, 129), (3, 70000)])
    def test_row_sum(self, shape):
        x = torch.randn(shape, dtype=torch.float32)
        out = linalg.coalesced_reduction(x)
        ref = x.double().sum(dim=1)
        torch.testing.assert_close(out.double(), ref, rtol=1e-5, atol=1e-5)

    def test_col_sum(self):
        x = torch.randn(64, 33)
        torch.testing.assert_close(linalg.strided_reduction(x).double(),
                                   x.double().sum(dim=0), rtol=1e-5, atol=1e-5)

    def test_unified_dispatch(self):
        x = torch.randn(10, 20)
        r = linalg.reduce(x, apply=Apply.ALONG_ROWS)
        c = linalg.reduce(x, apply=Apply.ALONG_COLUMNS)
        assert r.shape == (10,)
        assert c.shape == (20,)
        # col-major input flips the meaning
        r2 = linalg.reduce(x.t(), apply=Apply.ALONG_COLUMNS, row_major=False)
        torch.testing.assert_close(r, r2)

    @pytest.mark.parametrize("main_op,red", [("sq", "sum"), ("abs", "sum"),
                                             ("identity", "max"), ("identity", "min")])
    def test_ops(self, main_op, red):
        x = torch.randn(8, 100)
        out = linalg.coalesced_reduction(x, main_op=main_op, reduce_op=red)
        v = {"sq": x * x, "abs": x.abs(), "identity": x}[main_op]
        ref = {"sum": v.sum(dim=1), "max": v.max(dim=1).values,
               "min": v.min(dim=1).values}[red]
        torch.testing.assert_close(out, ref, rtol=1e-4, atol=1e-5)


class TestNorm:
    def test_row_norms(self):
        x = torch.randn(6, 40)
        torch.testing.assert_close(linalg.row_norm(x, NormType.L2),
                                   x.norm(dim=1), rtol=1e-5, atol=1e-6)
        torch.testing.assert_close(linalg.row_norm(x, NormType.L1),
                                   x.abs().sum(dim=1), rtol=1e-5, atol=1e-6)
        torch.testing.assert_close(linalg.row_norm(x, NormType.LINF),
                                   x.abs().max(dim=1).values)

    def test_normalize(self):
        x = torch.randn(5, 16)
        out = linalg.normalize(x)
        torch.testing.assert_close(out.norm(dim=1), torch.ones(5), rtol=1e-5, atol=1e-5)


class TestMap:
    def test_arith(self):
        x, y = torch.randn(10), torch.randn(10)
        torch.testing.assert_close(linalg.add(x, y), x + y)
        torch.testing.assert_close(linalg.map_then_reduce(lambda a, b: a * b, "sum", x, y),
                                   (x * y).sum())

    def test_map_offset(self):
        out = linalg.map_offset(lambda i: i.float() * 2, 5, device="cpu")
        torch.testing.assert_close(out, torch.tensor([0.0, 2, 4, 6, 8]))


class TestMatrixVector:
    def test_broadcast_rows_cols(self):
        m = torch.randn(4, 6)
        v = torch.randn(6)
        torch.testing.assert_close(linalg.matrix_vector_op(m, v, "add"), m + v)
        w = torch.randn(4)
        torch.testing.assert_close(linalg.matrix_vector_op(m, w, "mul", along_rows=False),
                                   m * w.unsqueeze(1))


class TestGemm:
    def test_gemm_gemv_dot_axpy(self):
        a, b = torch.randn(5, 4), torch.randn(4, 3)
        torch.testing.assert_close(linalg.gemm(a, b), a @ b)
        x = torch.randn(4)
        torch.testing.assert_close(linalg.gemv(a, x), a @ x)
        torch.testing.assert_close(linalg.dot(x, x), (x * x).sum())
        y = torch.zeros(4)
        linalg.axpy(2.0, x, y)
        torch.testing.assert_close(y, 2 * x)

    @pytest.mark.parametrize("mode,factor", [("bf16x2", 300.0), ("bf16x3", 6.0)])
    def test_fp32_emulation_accuracy(self, mode, factor):
        """Split-bf16 GEMM accuracy class vs fp64, relative to native fp32:
        bf16x3 must be fp32-class (within a small factor of native SGEMM
        rounding); bf16x2 is TF32-class (~2^-16)."""
        torch.manual_seed(0)
        a = torch.randn(64, 96)
        b = torch.randn(96, 48)
        ref = a.double() @ b.double()
        err_native = ((a @ b).double() - ref).abs().max()
        err = (linalg.gemm_fp32_emulated(a, b, mode=mode).double() - ref).abs().max()
        assert float(err) < float(err_native) * factor, \
            f"{mode} abs err {float(err)} vs native {float(err_native)}"

    def test_fp32_emulation_beats_bf16(self):
        torch.manual_seed(1)
        a, b = torch.randn(32, 64), torch.randn(64, 32)
        ref = a.double() @ b.double()
        bf = (a.bfloat16().float() @ b.bfloat16().float()).double()
        em = linalg.gemm_fp32_emulated(a, b, mode="bf16x3").double()
        assert (em - ref).abs().max() < (bf - ref).abs().max() / 50


class TestReduceByKey:
    def test_rows_by_key(self):
        x = torch.randn(20, 5)
        keys = torch.randint(0, 4, (20,))
        out = linalg.reduce_rows_by_key(x, keys, 4)
        ref = torch.zeros(4, 5)
        for i in range(20):
            ref[keys[i]] += x[i]
        torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-6)

    def test_cols_by_key(self):
        x = torch.randn(5, 12)
        keys = torch.randint(0, 3, (12,))
        out = linalg.reduce_cols_by_key(x, keys, 3)
        ref = torch.zeros(5, 3)
        for j in range(12):
            ref[:, keys[j]] += x[:, j]
        torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-6)


class TestDecomp:
    def test_eigh(self):
        a = torch.randn(8, 8, dtype=torch.float64)
        a = a + a.t()
        w, v = linalg.eigh(a)
        torch.testing.assert_close(v @ torch.diag(w) @ v.t(), a, rtol=1e-9, atol=1e-9)

    def test_svd_reconstruction(self):
        a = torch.randn(10, 6, dtype=torch.float64)
        u, s, v = linalg.svd(a)
        torch.testing.assert_close(u @ torch.diag(s) @ v.t(), a, rtol=1e-9, atol=1e-9)

    def test_qr(self):
        a = torch.randn(12, 5, dtype=torch.float64)
        q, r = linalg.qr(a)
        torch.testing.assert_close(q @ r, a, rtol=1e-9, atol=1e-9)
        torch.testing.assert_close(q.t() @ q, torch.eye(5, dtype=torch.float64),
                                   rtol=1e-8, atol=1e-8)

    def test_cholesky_r1_update(self):
        a = torch.randn(6, 6, dtype=torch.float64)
        a = a @ a.t() + 6 * torch.eye(6, dtype=torch.float64)
        x = torch.randn(6, dtype=torch.float64)
        l = linalg.cholesky(a)
        l2 = linalg.cholesky_r1_update(l, x)
        ref = linalg.cholesky(a + torch.outer(x, x))
        torch.testing.assert_close(l2, ref, rtol=1e-8, atol=1e-8)

    @pytest.mark.parametrize("algo", ["qr", "eig", "svd-qr"])
    def test_lstsq(self, algo):
        torch.manual_seed(0)
        a = torch.randn(50, 8, dtype=torch.float64)
        w_true = torch.randn(8, dtype=torch.float64)
        b = a @ w_true
        w = linalg.lstsq(a, b, algo=algo)
        torch.testing.assert_close(w, w_true, rtol=1e-6, atol=1e-6)


class TestRsvdPca:
    def test_rsvd_low_rank(self):
        torch.manual_seed(0)
        u0 = torch.randn(60, 5, dtype=torch.float64)
        v0 = torch.randn(5, 40, dtype=torch.float64)
        a = u0 @ v0
        u, s, v = linalg.rsvd(a, k=5, seed=0)
        torch.testing.assert_close(u @ torch.diag(s) @ v.t(), a, rtol=1e-6, atol=1e-6)

    def test_pca_matches_svd_path(self):
        torch.manual_seed(0)
        x = torch.randn(100, 10, dtype=torch.float64)
        m1 = linalg.pca_fit(x, 3, algo="eig")
        m2 = linalg.pca_fit(x, 3, algo="svd")
        torch.testing.assert_close(m1.explained_variance, m2.explained_variance,
                                   rtol=1e-6, atol=1e-8)
        # transform/inverse roundtrip recovers the projection
        z = linalg.pca_transform(m1, x)
        xr = linalg.pca_inverse_transform(m1, z)
        assert ((x - xr) ** 2).mean() < ((x - x.mean(0)) ** 2).mean()

    def test_tsvd(self):
        torch.manual_seed(0)
        x = torch.randn(40, 8, dtype=torch.float64)
        m = linalg.tsvd_fit(x, 3)
        assert m.components.shape == (3, 8)
        u, s, vh = torch.linalg.svd(x, full_matrices=False)
        torch.testing.assert_close(m.singular_values, s[:3], rtol=1e-8, atol=1e-8)


class TestDecompCrossValidation:
    def test_pca_matches_sklearn(self):
        from sklearn.decomposition import PCA
        from raft_amd.linalg import pca_fit, pca_transform
        torch.manual_seed(0)
        x = torch.randn(500, 12) @ torch.randn(12, 12)
        m = pca_fit(x, 4)
        sk = PCA(n_components=4).fit(x.numpy())
        # eigenvalue spectra agree; components up to sign
        torch.testing.assert_close(m.explained_variance.double(),
                                   torch.from_numpy(sk.explained_variance_).double(),
                                   rtol=1e-4, atol=1e-4)
        z = pca_transform(m, x).numpy()
        zs = sk.transform(x.numpy())
        import numpy as np
        for c in range(4):
            assert min(np.abs(z[:, c] - zs[:, c]).max(),
                       np.abs(z[:, c] + zs[:, c]).max()) < 1e-3

    def test_rsvd_matches_scipy_spectrum(self):
        import numpy as np
        from raft_amd.linalg import rsvd
        torch.manual_seed(1)
        a = torch.randn(300, 6) @ torch.randn(6, 200) + 0.01 * torch.randn(300, 200)
        _, s, _ = rsvd(a, k=6, n_iter=4, seed=0)
        ref = np.linalg.svd(a.numpy(), compute_uv=False)[:6]
        np.testing.assert_allclose(s.numpy(), ref, rtol=1e-3)


class TestEigVariants:
    """eig_jacobi is a REAL cyclic-Jacobi solver and eig_selective a real
    partial solver (syevj/syevdx parity, not eigh aliases)."""

    @pytest.mark.parametrize("n", [1, 2, 7, 32, 65])
    def test_jacobi_matches_eigh(self, n):
        torch.manual_seed(n)
        a = torch.randn(n, n, dtype=torch.float64)
        a = (a + a.t()) / 2
        w_j, v_j = eig_jacobi(a, tol=1e-12)
        w_r = torch.linalg.eigvalsh(a)
        torch.testing.assert_close(w_j, w_r, rtol=1e-8, atol=1e-8)
        # eigenvector residuals
        res = (a @ v_j - v_j * w_j.unsqueeze(0)).norm()
        assert float(res) < 1e-7 * max(1.0, float(a.norm()))
        # orthonormal
        torch.testing.assert_close(v_j.t() @ v_j, torch.eye(n, dtype=a.dtype),
                                   rtol=1e-8, atol=1e-8)

    def test_selective_lobpcg_path(self):
        torch.manual_seed(3)
        n, k = 256, 5
        a = torch.randn(n, n, dtype=torch.float64)
        a = (a + a.t()) / 2
        w_full = torch.linalg.eigvalsh(a)
        w_top, v_top = eig_selective(a, k, largest=True, method="lobpcg")
        torch.testing.assert_close(w_top, w_full[-k:], rtol=1e-6, atol=1e-6)
        w_bot, v_bot = eig_selective(a, k, largest=False, method="lobpcg")
        torch.testing.assert_close(w_bot, w_full[:k], rtol=1e-6, atol=1e-6)

    def test_selective_full_fallback(self):
        torch.manual_seed(4)
        a = torch.randn(16, 16)
        a = (a + a.t()) / 2
        w, v = eig_selective(a, 12, largest=True)   # wide range -> full
        torch.testing.assert_close(w, torch.linalg.eigvalsh(a)[-12:],
                                   rtol=1e-4, atol=1e-4)
