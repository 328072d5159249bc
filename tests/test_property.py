"""Property-based fuzzing (hypothesis) over the CPU reference paths.

The unit suites pin exact oracles; these check structural invariants over
randomized shapes/values — the reference's randomized-input test style
 extended with shrinking.
"""
import os

import pytest
import torch
from hypothesis import given, settings, strategies as st

# HYP_EXAMPLES=200 for an extended local fuzz; 25 keeps CI fast
settings.register_profile("ci", deadline=None,
                          max_examples=int(os.environ.get("HYP_EXAMPLES", "25")))
settings.load_profile("ci")


class TestSelectKProps:
    @given(b=st.integers(1, 8), n=st.integers(1, 200), kfrac=st.floats(0.01, 1.0),
           select_min=st.booleans())
    def test_values_and_indices_consistent(self, b, n, kfrac, select_min):
        from raft_amd.matrix import select_k
        k = max(1, int(n * kfrac))
        torch.manual_seed(0)
        x = torch.randn(b, n)
        vals, idx = select_k(x, k, select_min=select_min)
        ref = torch.sort(x, dim=1, descending=not select_min).values[:, :k]
        torch.testing.assert_close(vals, ref)
        torch.testing.assert_close(torch.gather(x, 1, idx), vals)
        for r in range(b):
            assert idx[r].unique().numel() == k


class TestRngProps:
    @given(seed=st.integers(0, 2**32 - 1), n=st.integers(1, 5000))
    def test_uniform_range_and_determinism(self, seed, n):
        from raft_amd.random import uniform, RngState
        a = uniform((n,), state=RngState(seed=seed))
        b = uniform((n,), state=RngState(seed=seed))
        assert torch.equal(a, b)
        assert (a >= 0).all() and (a < 1).all()

    @given(seed=st.integers(0, 2**31), n=st.integers(2, 300), k=st.integers(1, 100))
    def test_sample_without_replacement_unique(self, seed, n, k):
        from raft_amd.random import sample_without_replacement, RngState
        k = min(k, n)
        idx = sample_without_replacement(n, k, state=RngState(seed=seed))
        assert idx.unique().numel() == k
        assert (idx >= 0).all() and (idx < n).all()


class TestSparseProps:
    @given(n=st.integers(2, 40), density=st.floats(0.05, 0.6),
           seed=st.integers(0, 1000))
    def test_coo_csr_roundtrip(self, n, density, seed):
        from raft_amd.sparse import COO, coo_to_csr, csr_to_coo
        torch.manual_seed(seed)
        dense = torch.randn(n, n) * (torch.rand(n, n) < density)
        nz = dense.nonzero(as_tuple=False)
        coo = COO(nz[:, 0].to(torch.int32), nz[:, 1].to(torch.int32),
                  dense[nz[:, 0], nz[:, 1]], n, n)
        back = csr_to_coo(coo_to_csr(coo))
        rebuilt = torch.zeros_like(dense)
        rebuilt[back.rows.long(), back.cols.long()] = back.values
        torch.testing.assert_close(rebuilt, dense)

    @given(n=st.integers(2, 30), seed=st.integers(0, 1000))
    def test_laplacian_row_sums_zero(self, n, seed):
        from raft_amd.sparse import COO, coo_to_csr, symmetrize_coo, laplacian
        torch.manual_seed(seed)
        dense = (torch.rand(n, n) < 0.3).float()
        dense.fill_diagonal_(0)
        nz = dense.nonzero(as_tuple=False)
        if nz.numel() == 0:
            return
        coo = COO(nz[:, 0].to(torch.int32), nz[:, 1].to(torch.int32),
                  torch.ones(nz.shape[0]), n, n)
        lap = laplacian(coo_to_csr(symmetrize_coo(coo)))
        row_sums = lap.to_torch_sparse().to_dense().sum(dim=1)
        torch.testing.assert_close(row_sums, torch.zeros(n), atol=1e-5, rtol=0)


class TestLabelProps:
    @given(n=st.integers(1, 200), hi=st.integers(1, 50), seed=st.integers(0, 999))
    def test_make_monotonic_preserves_partition(self, n, hi, seed):
        from raft_amd.label import make_monotonic
        torch.manual_seed(seed)
        labels = torch.randint(0, hi, (n,))
        mono = make_monotonic(labels)
        k = labels.unique().numel()
        assert mono.min() == 0 and mono.max() == k - 1
        # same partition: pairwise equality preserved
        eq_orig = labels.unsqueeze(0) == labels.unsqueeze(1)
        eq_mono = mono.unsqueeze(0) == mono.unsqueeze(1)
        assert torch.equal(eq_orig, eq_mono)


class TestLapProps:
    @given(n=st.integers(2, 12), seed=st.integers(0, 500))
    def test_auction_matches_scipy_optimum(self, n, seed):
        from raft_amd.solver import linear_assignment
        from scipy.optimize import linear_sum_assignment
        torch.manual_seed(seed)
        cost = torch.rand(n, n) * 10
        assign, my_cost = linear_assignment(cost)
        assert assign.long().unique().numel() == n  # a permutation
        cd = cost.double().numpy()       # compare in fp64 (fp32 ties differ)
        ri, ci = linear_sum_assignment(cd)
        opt = float(cd[ri, ci].sum())
        # eps scales down to tol/n (absolute), so the gap is ~1e-8-tiny
        assert my_cost <= opt + 1e-6


class TestReduceByKeyProps:
    @given(n=st.integers(1, 500), d=st.integers(1, 8), k=st.integers(1, 16),
           seed=st.integers(0, 999), weighted=st.booleans())
    def test_matches_scatter_add(self, n, d, k, seed, weighted):
        from raft_amd.linalg import reduce_rows_by_key
        torch.manual_seed(seed)
        x = torch.randn(n, d)
        keys = torch.randint(0, k, (n,), dtype=torch.int32)
        w = torch.rand(n) if weighted else None
        out = reduce_rows_by_key(x, keys, n_keys=k, weights=w)
        ref = torch.zeros(k, d)
        xw = x if w is None else x * w.unsqueeze(1)
        ref.index_add_(0, keys.long(), xw)
        torch.testing.assert_close(out, ref, atol=1e-5, rtol=1e-5)


class TestMatrixProps:
    @given(r=st.integers(1, 20), c=st.integers(1, 20), k=st.integers(0, 25))
    def test_shift_reverse_involutions(self, r, c, k):
        from raft_amd import matrix
        x = torch.randn(r, c)
        assert torch.equal(matrix.row_reverse(matrix.row_reverse(x)), x)
        assert torch.equal(matrix.col_reverse(matrix.col_reverse(x)), x)
        sh = matrix.shift_rows(x, k % (r + 1), fill_value=0.0)
        assert sh.shape == x.shape


class TestStatsProps:
    @given(n=st.integers(2, 400), d=st.integers(1, 10), seed=st.integers(0, 999))
    def test_meanvar_matches_torch(self, n, d, seed):
        from raft_amd.stats import meanvar
        torch.manual_seed(seed)
        x = torch.randn(n, d) * 3 + 1
        mu, var = meanvar(x)
        torch.testing.assert_close(mu.double(), x.double().mean(0), atol=1e-5, rtol=1e-5)
        torch.testing.assert_close(var.double(), x.double().var(0), atol=1e-4, rtol=1e-4)

    @given(n=st.integers(1, 2000), bins=st.integers(1, 64), seed=st.integers(0, 999))
    def test_histogram_total_and_support(self, n, bins, seed):
        from raft_amd.stats import histogram
        torch.manual_seed(seed)
        x = torch.randn(n)
        h = histogram(x, n_bins=bins)
        assert int(h.sum()) == n
        assert (h >= 0).all()

    @given(n=st.integers(1, 300), k=st.integers(1, 8), seed=st.integers(0, 99))
    def test_contingency_marginals(self, n, k, seed):
        from raft_amd.stats import contingency_matrix
        torch.manual_seed(seed)
        a = torch.randint(0, k, (n,))
        b = torch.randint(0, k, (n,))
        c = contingency_matrix(a, b)
        assert int(c.sum()) == n
        # rows span the label RANGE a.min()..a.max() (reference's
        # label-range reduction), so marginals are the shifted bincount
        torch.testing.assert_close(c.sum(dim=1),
                                   torch.bincount(a - a.min(),
                                                  minlength=c.shape[0]))


class TestBitsetProps:
    @given(n=st.integers(1, 500), seed=st.integers(0, 999))
    def test_set_test_flip_count(self, n, seed):
        from raft_amd.core import Bitset
        torch.manual_seed(seed)
        bs = Bitset(n, default=False)  # reference default is all-set
        idx = torch.randperm(n)[: max(1, n // 3)]
        bs.set(idx)
        assert bs.count() == idx.numel()
        assert bool(bs.test(idx).all())
        bs.flip()
        assert bs.count() == n - idx.numel()


class TestNormalizeProps:
    @given(n=st.integers(1, 100), d=st.integers(1, 50), seed=st.integers(0, 999))
    def test_row_normalize_unit_norm(self, n, d, seed):
        from raft_amd.linalg import normalize
        torch.manual_seed(seed)
        x = torch.randn(n, d) * 10
        y = normalize(x)
        norms = y.norm(dim=1)
        nz = x.norm(dim=1) > 1e-6
        torch.testing.assert_close(norms[nz], torch.ones(int(nz.sum())),
                                   atol=1e-4, rtol=1e-4)


class TestSerializeProps:
    @given(r=st.integers(0, 20), c=st.integers(1, 20),
           dt=st.sampled_from(["float32", "float64", "int32", "int64", "uint8"]),
           seed=st.integers(0, 999))
    def test_npy_roundtrip(self, r, c, dt, seed):
        import io
        from raft_amd.core import serialize_mdspan, deserialize_mdspan
        torch.manual_seed(seed)
        dtype = getattr(torch, dt)
        if dtype.is_floating_point:
            x = torch.randn(r, c).to(dtype)
        else:
            x = torch.randint(0, 100, (r, c), dtype=dtype)
        buf = io.BytesIO()
        serialize_mdspan(buf, x)
        buf.seek(0)
        y = deserialize_mdspan(buf)
        assert y.dtype == x.dtype and torch.equal(y, x)

    @given(r=st.integers(1, 16), c=st.integers(1, 16), seed=st.integers(0, 99))
    def test_npy_numpy_interchange(self, r, c, seed):
        """Our .npy bytes load in NumPy and vice versa (format parity)."""
        import io
        import numpy as np
        from raft_amd.core import serialize_mdspan, deserialize_mdspan
        torch.manual_seed(seed)
        x = torch.randn(r, c)
        buf = io.BytesIO()
        serialize_mdspan(buf, x)
        buf.seek(0)
        arr = np.load(buf)
        assert np.array_equal(arr, x.numpy())
        buf2 = io.BytesIO()
        np.save(buf2, arr)
        buf2.seek(0)
        y = deserialize_mdspan(buf2)
        assert torch.equal(y, x)
