import numpy as np
import pytest
import scipy.sparse as sp
import scipy.sparse.linalg as spla
import torch

from raft_amd import sparse as rsp
from raft_amd.sparse import CSR, COO
from raft_amd.sparse.solver import eigsh, mst, randomized_svds


def _rand_csr(m, n, density=0.1, seed=0, symmetric=False):
    rng = np.random.RandomState(seed)
    s = sp.random(m, n, density=density, random_state=rng, format="csr", dtype=np.float64)
    if symmetric:
        s = (s + s.T) * 0.5
        s = s.tocsr()
    return s


def _to_csr(s) -> CSR:
    return CSR(torch.from_numpy(s.indptr.astype(np.int64)),
               torch.from_numpy(s.indices.astype(np.int64)),
               torch.from_numpy(s.data), s.shape[0], s.shape[1])


class TestConvert:
    def test_coo_csr_roundtrip(self):
        s = _rand_csr(20, 15)
        csr = _to_csr(s)
        coo = rsp.csr_to_coo(csr)
        csr2 = rsp.coo_to_csr(coo)
        np.testing.assert_array_equal(csr2.indptr.numpy(), s.indptr)
        np.testing.assert_array_equal(csr2.indices.numpy(), s.indices)
        np.testing.assert_allclose(csr2.values.numpy(), s.data)

    def test_dense_roundtrip(self):
        d = torch.randn(6, 9)
        d[d.abs() < 0.7] = 0
        csr = rsp.dense_to_csr(d)
        torch.testing.assert_close(rsp.csr_to_dense(csr), d)

    def test_adj_to_csr(self):
        adj = torch.rand(5, 5) > 0.5
        csr = rsp.adj_to_csr(adj)
        torch.testing.assert_close(rsp.csr_to_dense(csr), adj.float())

    def test_bitmap_to_csr(self):
        from raft_amd.core import Bitset
        dense = torch.rand(24) > 0.5
        bs = Bitset.from_dense(dense)
        csr = rsp.bitmap_to_csr(bs, 4, 6)
        torch.testing.assert_close(rsp.csr_to_dense(csr),
                                   dense.reshape(4, 6).float())


class TestLinalg:
    def test_spmv_vs_scipy(self):
        s = _rand_csr(30, 20)
        x = np.random.rand(20)
        y = rsp.spmv(_to_csr(s), torch.from_numpy(x))
        np.testing.assert_allclose(y.numpy(), s @ x, rtol=1e-12)

    def test_spmm(self):
        s = _rand_csr(10, 8)
        b = np.random.rand(8, 3)
        c = rsp.spmm(_to_csr(s), torch.from_numpy(b))
        np.testing.assert_allclose(c.numpy(), s @ b, rtol=1e-12)

    def test_sddmm(self):
        s = _rand_csr(6, 7, density=0.4)
        a = np.random.rand(6, 4)
        b = np.random.rand(7, 4)
        out = rsp.sddmm(torch.from_numpy(a), torch.from_numpy(b), _to_csr(s))
        ref = (a @ b.T)[s.nonzero()]
        np.testing.assert_allclose(out.values.numpy(), ref, rtol=1e-12)

    def test_laplacian(self):
        s = _rand_csr(12, 12, density=0.3, symmetric=True)
        s.setdiag(0)
        s.eliminate_zeros()
        lap = rsp.laplacian(_to_csr(s))
        ref = sp.csgraph.laplacian(s)
        np.testing.assert_allclose(rsp.csr_to_dense(lap).numpy(), ref.toarray(),
                                   rtol=1e-10, atol=1e-12)

    def test_symmetrize(self):
        coo_s = sp.random(8, 8, density=0.3, random_state=np.random.RandomState(1),
                          format="coo", dtype=np.float64)
        coo = COO(torch.from_numpy(coo_s.row.astype(np.int64)),
                  torch.from_numpy(coo_s.col.astype(np.int64)),
                  torch.from_numpy(coo_s.data), 8, 8)
        sym = rsp.symmetrize_coo(coo, op="add")
        ref = (coo_s + coo_s.T).toarray()
        np.testing.assert_allclose(sym.to_dense().numpy(), ref, rtol=1e-12)

    def test_transpose_norm_degree_add(self):
        s = _rand_csr(9, 5)
        t = rsp.csr_transpose(_to_csr(s))
        np.testing.assert_allclose(rsp.csr_to_dense(t).numpy(), s.T.toarray())
        rn = rsp.csr_row_norm(_to_csr(s), "l2")
        np.testing.assert_allclose(rn.numpy(), np.asarray((s.multiply(s)).sum(axis=1)).ravel(),
                                   rtol=1e-12)
        deg = rsp.csr_degree(_to_csr(s))
        np.testing.assert_array_equal(deg.numpy(), np.diff(s.indptr))
        s2 = _rand_csr(9, 5, seed=3)
        add = rsp.csr_add(_to_csr(s), _to_csr(s2))
        np.testing.assert_allclose(rsp.csr_to_dense(add).numpy(), (s + s2).toarray(),
                                   rtol=1e-12)


class TestOps:
    def test_filter_and_dedupe(self):
        rows = torch.tensor([0, 0, 1, 1, 1])
        cols = torch.tensor([1, 1, 0, 2, 2])
        vals = torch.tensor([1.0, 3.0, 0.0, 2.0, 5.0])
        coo = COO(rows, cols, vals, 2, 3)
        f = rsp.filter_zeros(coo)
        assert f.nnz == 4
        d = rsp.dedupe_coo(coo, op="max")
        dense = d.to_dense()
        assert dense[0, 1] == 3.0 and dense[1, 2] == 5.0

    def test_slice_rows(self):
        s = _rand_csr(10, 6)
        sl = rsp.slice_csr_rows(_to_csr(s), 2, 7)
        np.testing.assert_allclose(rsp.csr_to_dense(sl).numpy(), s[2:7].toarray())


class TestSelectK:
    def test_csr_select_k(self):
        s = _rand_csr(8, 30, density=0.5)
        vals, cols = rsp.csr_select_k(_to_csr(s), k=3, select_min=False)
        dense = torch.from_numpy(s.toarray())
        dense_masked = torch.where(dense != 0, dense, torch.tensor(float("-inf"), dtype=dense.dtype))
        ref = torch.topk(dense_masked, 3, dim=1)
        finite = torch.isfinite(ref.values)
        torch.testing.assert_close(vals[finite], ref.values[finite])


class TestPreprocessing:
    def test_tfidf(self):
        s = _rand_csr(10, 12, density=0.4)
        s.data = np.abs(s.data) + 1
        out = rsp.tfidf_transform(_to_csr(s))
        assert out.nnz == s.nnz
        assert (out.values > 0).all()

    def test_bm25(self):
        s = _rand_csr(10, 12, density=0.4)
        s.data = np.abs(s.data) + 1
        out = rsp.bm25_transform(_to_csr(s))
        assert out.nnz == s.nnz
        assert torch.isfinite(out.values).all()


class TestLanczos:
    def test_smallest_eigs_vs_scipy(self):
        s = _rand_csr(120, 120, density=0.05, seed=2, symmetric=True)
        s = s + sp.eye(120) * 0.1
        w_ref = np.sort(spla.eigsh(s, k=4, which="SA", maxiter=5000)[0])
        w, v = eigsh(_to_csr(s), k=4, tol=1e-10, seed=1)
        np.testing.assert_allclose(w.numpy(), w_ref, rtol=1e-5, atol=1e-7)
        # residual check ||Av - wv||
        a = torch.from_numpy(s.toarray())
        res = (a @ v - v * w.unsqueeze(0)).norm(dim=0)
        assert float(res.max()) < 1e-5

    def test_dense_operator(self):
        a = torch.randn(60, 60, dtype=torch.float64)
        a = a + a.t()
        w, v = eigsh(a, k=3, tol=1e-10)
        w_ref = torch.linalg.eigvalsh(a)[:3]
        torch.testing.assert_close(w, w_ref, rtol=1e-6, atol=1e-8)


class TestMST:
    def test_vs_scipy(self):
        s = _rand_csr(40, 40, density=0.2, seed=5, symmetric=True)
        s.setdiag(0)
        s.eliminate_zeros()
        s.data = np.abs(s.data) + 0.01
        ref = sp.csgraph.minimum_spanning_tree(s)
        src, dst, w = mst(_to_csr(s))
        # same total weight and edge count (MST may differ on ties; weights unique here)
        assert abs(float(w.sum()) - ref.sum()) < 1e-6
        assert src.numel() == ref.nnz


class TestRandomizedSvds:
    def test_low_rank_recovery(self):
        rng = np.random.RandomState(0)
        u0 = rng.rand(50, 4)
        v0 = rng.rand(4, 30)
        dense = u0 @ v0            # exactly rank 4 (stored sparse for the path)
        s = sp.csr_matrix(dense)
        csr = _to_csr(s)
        u, sv, v = randomized_svds(csr, k=4, n_iter=6, seed=0)
        approx = (u * sv.unsqueeze(0)) @ v.t()
        np.testing.assert_allclose(approx.numpy(), dense, atol=1e-6)


class TestCsrDiagonal:
    def test_extract_and_set(self):
        from raft_amd.sparse import CSR, csr_diagonal, csr_set_diagonal
        d = torch.tensor([[1.0, 2.0, 0.0],
                          [0.0, 0.0, 3.0],
                          [4.0, 0.0, 5.0]])
        a = CSR.from_dense(d)
        torch.testing.assert_close(csr_diagonal(a), torch.tensor([1.0, 0.0, 5.0]))
        b = csr_set_diagonal(a, torch.tensor([9.0, 8.0, 7.0]))
        bd = b.to_torch_sparse().to_dense()
        assert bd[0, 0] == 9.0 and bd[2, 2] == 7.0
        assert bd[1, 1] == 0.0  # not stored -> pattern unchanged
        assert bd[0, 1] == 2.0


class TestCooSpmv:
    def test_spmv_spmm_accept_coo(self):
        from raft_amd.sparse import COO, spmv, spmm
        d = torch.tensor([[1.0, 0.0, 2.0], [0.0, 3.0, 0.0], [4.0, 0.0, 0.0]])
        nz = d.nonzero(as_tuple=False)
        # deliberately unsorted triplets
        perm = torch.tensor([2, 0, 3, 1])
        coo = COO(nz[perm, 0].to(torch.int32), nz[perm, 1].to(torch.int32),
                  d[nz[perm, 0], nz[perm, 1]], 3, 3)
        x = torch.tensor([1.0, 2.0, 3.0])
        torch.testing.assert_close(spmv(coo, x), d @ x)
        b = torch.randn(3, 4)
        torch.testing.assert_close(spmm(coo, b), d @ b)


class TestLaplacianVsScipy:
    def test_laplacian_matches_scipy(self):
        import numpy as np
        import scipy.sparse as sp
        from scipy.sparse.csgraph import laplacian as sp_lap
        from raft_amd.sparse import CSR, laplacian, laplacian_normalized
        torch.manual_seed(0)
        dense = (torch.rand(40, 40) < 0.2).float()
        dense = ((dense + dense.T) > 0).float()
        dense.fill_diagonal_(0)
        a = CSR.from_dense(dense)
        ref = sp_lap(sp.csr_matrix(dense.numpy())).toarray()
        torch.testing.assert_close(laplacian(a).to_torch_sparse().to_dense(),
                                   torch.from_numpy(ref).float(),
                                   atol=1e-5, rtol=1e-5)
        refn = sp_lap(sp.csr_matrix(dense.numpy()), normed=True).toarray()
        torch.testing.assert_close(
            laplacian_normalized(a).to_torch_sparse().to_dense(),
            torch.from_numpy(refn).float(), atol=1e-5, rtol=1e-5)
