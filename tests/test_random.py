import math

import pytest
import torch

from raft_amd import random as rnd
from raft_amd.random import RngState


class TestRng:
    def test_uniform_reproducible(self):
        a = rnd.uniform((1000,), state=RngState(seed=7))
        b = rnd.uniform((1000,), state=RngState(seed=7))
        assert torch.equal(a, b)
        c = rnd.uniform((1000,), state=RngState(seed=8))
        assert not torch.equal(a, c)

    def test_uniform_moments(self):
        u = rnd.uniform((200000,), state=RngState(seed=1))
        assert abs(u.mean().item() - 0.5) < 0.01
        assert abs(u.var().item() - 1.0 / 12) < 0.005
        assert u.min() >= 0 and u.max() <= 1

    def test_normal_moments(self):
        z = rnd.normal((200000,), mu=2.0, sigma=3.0, state=RngState(seed=2))
        assert abs(z.mean().item() - 2.0) < 0.05
        assert abs(z.std().item() - 3.0) < 0.05

    @pytest.mark.parametrize("fn,kw,mean,var", [
        (rnd.exponential, {"lambda_": 2.0}, 0.5, 0.25),
        (rnd.rayleigh, {"sigma": 1.0}, math.sqrt(math.pi / 2), (4 - math.pi) / 2),
        (rnd.laplace, {"mu": 0.0, "scale": 1.0}, 0.0, 2.0),
        (rnd.gumbel, {"mu": 0.0, "beta": 1.0}, 0.5772, math.pi ** 2 / 6),
        (rnd.logistic, {"mu": 0.0, "scale": 1.0}, 0.0, math.pi ** 2 / 3),
    ])
    def test_distribution_moments(self, fn, kw, mean, var):
        x = fn((300000,), state=RngState(seed=3), **kw).double()
        assert abs(x.mean().item() - mean) < 0.03 * max(1, abs(mean))
        assert abs(x.var().item() - var) < 0.05 * max(1.0, var)

    def test_bernoulli(self):
        b = rnd.bernoulli((100000,), p=0.3, state=RngState(seed=4))
        assert abs(b.float().mean().item() - 0.3) < 0.01

    def test_lognormal(self):
        x = rnd.lognormal((200000,), state=RngState(seed=5)).double()
        ref_mean = math.exp(0.5)
        assert abs(x.mean().item() - ref_mean) < 0.05 * ref_mean

    def test_sample_without_replacement_unique(self):
        idx = rnd.sample_without_replacement(100, 30, state=RngState(seed=6))
        assert idx.unique().numel() == 30
        assert idx.max() < 100

    def test_sample_with_replacement_weighted(self):
        w = torch.tensor([0.0, 1.0, 3.0])
        s = rnd.sample_with_replacement(w, 40000, state=RngState(seed=7))
        counts = torch.bincount(s, minlength=3).float()
        assert counts[0] == 0
        assert abs(counts[2] / counts[1] - 3.0) < 0.2


class TestMakeBlobs:
    def test_shapes_and_labels(self):
        x, y, c = rnd.make_blobs(500, 8, n_clusters=4, state=RngState(seed=0))
        assert x.shape == (500, 8) and y.shape == (500,) and c.shape == (4, 8)
        assert y.min() >= 0 and y.max() < 4

    def test_points_near_centers(self):
        x, y, c = rnd.make_blobs(2000, 4, n_clusters=3, cluster_std=0.2,
                                 center_box=(-20, 20), state=RngState(seed=1))
        d = (x - c[y]).norm(dim=1)
        assert d.mean() < 0.2 * math.sqrt(4) * 2


class TestMakeRegression:
    def test_linear_model_recoverable(self):
        x, y, coef = rnd.make_regression(300, 10, noise=0.0, state=RngState(seed=0))
        w = torch.linalg.lstsq(x.double(), y.double().unsqueeze(1)).solution.squeeze(1)
        torch.testing.assert_close(w.float(), coef.squeeze(1), rtol=1e-3, atol=1e-3)

    def test_effective_rank(self):
        x, y, _ = rnd.make_regression(100, 30, effective_rank=3, tail_strength=0.01,
                                      state=RngState(seed=0))
        s = torch.linalg.svdvals(x.double())
        assert s[8] < s[0] * 0.1  # spectrum decays fast


class TestRmat:
    def test_bounds_and_skew(self):
        src, dst = rnd.rmat(8, 8, 20000, a=0.7, b=0.1, c=0.1, state=RngState(seed=0))
        assert src.max() < 256 and dst.max() < 256
        assert src.min() >= 0 and dst.min() >= 0
        # skewed quadrant probabilities put most mass at low ids
        assert (src < 128).float().mean() > 0.7

    def test_bit_recursion_matches_theta(self):
        """Each of the r_scale recursion levels independently picks the
        row-half with P(top) = a + b: the per-LEVEL frequency must match,
        not just the top-level split."""
        a, b, c = 0.6, 0.15, 0.15
        src, dst = rnd.rmat(10, 10, 100000, a=a, b=b, c=c, state=RngState(seed=3))
        for level in range(10):
            bit = (src >> (9 - level)) & 1
            p_top = float((bit == 0).float().mean())
            assert abs(p_top - (a + b)) < 0.02, (level, p_top)


class TestPermute:
    def test_is_permutation(self):
        p = rnd.permute(1000, state=RngState(seed=0))
        assert torch.equal(torch.sort(p).values, torch.arange(1000))


class TestMvg:
    def test_covariance_recovery(self):
        mean = torch.tensor([1.0, -2.0])
        cov = torch.tensor([[2.0, 0.6], [0.6, 1.0]])
        s = rnd.multi_variable_gaussian(mean, cov, 100000, state=RngState(seed=0))
        emp_mean = s.mean(dim=0)
        emp_cov = torch.cov(s.t())
        torch.testing.assert_close(emp_mean, mean, rtol=0.05, atol=0.05)
        torch.testing.assert_close(emp_cov, cov, rtol=0.08, atol=0.08)


class TestDistributionShapes:
    """Kolmogorov-Smirnov vs scipy's exact CDFs (beyond the moment checks):
    catches shape errors moments can't (e.g. a wrong tail transform)."""

    @pytest.mark.parametrize("name,fn_kwargs,scipy_dist", [
        ("uniform", {}, ("uniform", ())),
        ("normal", {}, ("norm", ())),
        ("exponential", {"lambda_": 1.0}, ("expon", ())),
        ("rayleigh", {"sigma": 1.0}, ("rayleigh", ())),
        ("laplace", {"mu": 0.0, "scale": 1.0}, ("laplace", ())),
        ("gumbel", {"mu": 0.0, "beta": 1.0}, ("gumbel_r", ())),
        ("logistic", {"mu": 0.0, "scale": 1.0}, ("logistic", ())),
        ("lognormal", {"mu": 0.0, "sigma": 1.0}, ("lognorm", (1.0,))),
    ])
    def test_ks_vs_scipy(self, name, fn_kwargs, scipy_dist):
        import scipy.stats as ss
        fn = getattr(rnd, name)
        x = fn((100000,), state=RngState(seed=11), **fn_kwargs).double().numpy()
        dist_name, args = scipy_dist
        stat, pvalue = ss.kstest(x, dist_name, args=args)
        assert pvalue > 1e-4, f"{name}: KS stat {stat:.4f}, p {pvalue:.2e}"


class TestMvgDecomposers:
    @pytest.mark.parametrize("method", ["chol", "eig", "qr"])
    def test_all_decomposers_recover_cov(self, method):
        mean = torch.tensor([1.0, -2.0, 0.5])
        cov = torch.tensor([[2.0, 0.5, 0.1], [0.5, 1.0, 0.2], [0.1, 0.2, 1.5]])
        s = rnd.multi_variable_gaussian(mean, cov, 200_000, method=method,
                                        state=RngState(seed=6))
        torch.testing.assert_close(s.mean(dim=0), mean, atol=0.03, rtol=0)
        torch.testing.assert_close(torch.cov(s.t()), cov, atol=0.05, rtol=0.05)


class TestPhilox:
    """Philox4x32-10 generator (reference rng_device.cuh PhiloxGenerator:426)."""

    def test_distribution_moments(self):
        from raft_amd.random.rng import RngState, uniform, normal
        s = RngState(seed=5, gen_type="philox")
        u = uniform((200000,), state=s)
        assert abs(float(u.mean()) - 0.5) < 3e-3
        assert abs(float(u.var()) - 1 / 12) < 2e-3
        z = normal((200000,), state=s)
        assert abs(float(z.mean())) < 8e-3
        assert abs(float(z.var()) - 1.0) < 2e-2

    def test_counter_based_reproducible(self):
        from raft_amd.random.rng import RngState, uniform
        a = uniform((1000,), state=RngState(seed=9, gen_type="philox"))
        b = uniform((1000,), state=RngState(seed=9, gen_type="philox"))
        assert torch.equal(a, b)
        c = uniform((1000,), state=RngState(seed=10, gen_type="philox"))
        assert not torch.equal(a, c)

    def test_differs_from_pcg(self):
        from raft_amd.random.rng import RngState, uniform
        a = uniform((1000,), state=RngState(seed=9, gen_type="philox"))
        b = uniform((1000,), state=RngState(seed=9, gen_type="pcg"))
        assert not torch.equal(a, b)

    def test_known_vector(self):
        # Philox4x32-10 reference vector (counter=0, key=0): x0 = 0x6627e8d5
        from raft_amd.random.rng import _philox_block
        x0 = int(_philox_block(0, 0, torch.zeros(1, dtype=torch.int64))[0])
        assert x0 == 0x6627E8D5, hex(x0)
