"""Memory-resource layer tests.

Reference parity: rmm pool/limiting/tracking resource semantics assumed by
raft primitives (core/resource/resource_types.hpp:37-40, mr/*.hpp,
memory_stats_resources.hpp:75). Everything here runs on CPU tensors — the
semantics are device-agnostic.
"""
import pytest
import torch

from raft_amd.core import (
    LimitingAdaptor,
    MemoryLimitExceeded,
    PoolMemoryResource,
    Resources,
    TorchMemoryResource,
    TrackingAdaptor,
)


class TestPool:
    def test_reuse_same_block(self):
        mr = PoolMemoryResource(initial_pool_size=1 << 16, device="cpu")
        with mr.allocate(1024) as a:
            pa = a.tensor.data_ptr()
        with mr.allocate(1024) as b:
            assert b.tensor.data_ptr() == pa  # freed block is recycled
        assert mr.outstanding_bytes() == 0

    def test_alignment(self):
        mr = PoolMemoryResource(initial_pool_size=1 << 16, device="cpu")
        with mr.allocate(1) as a, mr.allocate(1) as b:
            assert abs(b.tensor.data_ptr() - a.tensor.data_ptr()) >= 256

    def test_coalescing(self):
        mr = PoolMemoryResource(initial_pool_size=1 << 16, device="cpu")
        a = mr.allocate(4096)
        b = mr.allocate(4096)
        c = mr.allocate(4096)
        a.free()
        c.free()
        b.free()  # middle free must coalesce back into one run
        big = mr.allocate((1 << 16) - 256)
        big.free()

    def test_growth_and_maximum(self):
        mr = PoolMemoryResource(initial_pool_size=1 << 12,
                                maximum_pool_size=1 << 14, device="cpu")
        bufs = [mr.allocate(1 << 12) for _ in range(4)]  # grows to the max
        assert mr.pool_bytes() <= 1 << 14
        with pytest.raises(MemoryLimitExceeded):
            mr.allocate(1 << 12)
        for b in bufs:
            b.free()
        big = mr.allocate(1 << 13)  # fits after frees
        big.free()

    def test_oversize_single_alloc_raises(self):
        mr = PoolMemoryResource(initial_pool_size=1 << 12,
                                maximum_pool_size=1 << 13, device="cpu")
        with pytest.raises(MemoryLimitExceeded):
            mr.allocate(1 << 20)

    def test_view_shapes_dtypes(self):
        mr = PoolMemoryResource(initial_pool_size=1 << 16, device="cpu")
        with mr.allocate_tensor((8, 16), torch.float32) as w:
            t = w.view((8, 16), torch.float32)
            assert t.shape == (8, 16) and t.dtype == torch.float32
            t.fill_(3.0)
            assert float(t.sum()) == 3.0 * 128


class TestAdaptors:
    def test_limiting(self):
        mr = LimitingAdaptor(TorchMemoryResource("cpu"), limit_bytes=4096)
        a = mr.allocate(2048)
        with pytest.raises(MemoryLimitExceeded):
            mr.allocate(4096)
        a.free()
        b = mr.allocate(4096)  # fits once outstanding returns to 0
        b.free()
        assert mr.outstanding_bytes() == 0

    def test_tracking_counts_and_peak(self):
        mr = TrackingAdaptor(TorchMemoryResource("cpu"))
        a = mr.allocate(1000)
        b = mr.allocate(500)
        a.free()
        c = mr.allocate(200)
        s = mr.stats
        assert s.alloc_count == 3 and s.free_count == 1
        assert s.peak_bytes == 1500
        assert s.outstanding_bytes == 700
        with pytest.raises(RuntimeError):
            mr.assert_no_leaks()
        b.free()
        c.free()
        mr.assert_no_leaks()

    def test_stacked_chain(self):
        # tracking over limiting over pool — the full reference adaptor stack
        pool = PoolMemoryResource(initial_pool_size=1 << 14, device="cpu")
        mr = TrackingAdaptor(LimitingAdaptor(pool, limit_bytes=1 << 13))
        with mr.allocate(4096):
            assert mr.stats.outstanding_bytes == 4096
            with pytest.raises(MemoryLimitExceeded):
                mr.allocate(1 << 13)
        mr.assert_no_leaks()


class TestResourcesIntegration:
    def test_workspace_through_resources(self):
        res = Resources(device="cpu")
        mr = TrackingAdaptor(PoolMemoryResource(initial_pool_size=1 << 16,
                                                device="cpu"))
        res.set_workspace_resource(mr)
        with res.get_workspace((32, 32), torch.float32) as w:
            t = w.view((32, 32), torch.float32)
            t.zero_()
        mr.assert_no_leaks()
        assert mr.stats.alloc_count == 1

    def test_workspace_limit_enforced(self):
        res = Resources(device="cpu")
        res.set_workspace_resource(TorchMemoryResource("cpu"))
        res.set_workspace_limit(1 << 12)
        with pytest.raises(MemoryLimitExceeded):
            res.get_workspace((1 << 13,), torch.uint8)
        assert res.workspace_budget() == 1 << 12

    def test_budget_tracks_outstanding(self):
        res = Resources(device="cpu")
        res.set_workspace_resource(TorchMemoryResource("cpu"))
        res.set_workspace_limit(10000)
        w = res.get_workspace((4000,), torch.uint8)
        assert res.workspace_budget() == 6000
        w.free()
        assert res.workspace_budget() == 10000


class TestKnnBudgetedChunking:
    def test_tiles_shrink_with_budget(self):
        from raft_amd.neighbors.brute_force import _tiles_from_budget
        qc_big, ic_big = _tiles_from_budget(1 << 34, 16384, 10_000_000, 128, 64)
        qc_small, ic_small = _tiles_from_budget(1 << 24, 16384, 10_000_000, 128, 64)
        assert ic_small < ic_big
        assert ic_small >= 4 * 64  # never below the select floor
        # tile fits in half the budget
        assert qc_small * ic_small * 4 <= (1 << 24)

    def test_knn_with_capped_workspace(self):
        # CPU tiled path sized from an enforced 8 MiB workspace cap
        from raft_amd.neighbors import knn
        res = Resources(device="cpu")
        res.set_workspace_resource(TorchMemoryResource("cpu"))
        res.set_workspace_limit(8 << 20)
        torch.manual_seed(0)
        x = torch.randn(5000, 32)
        q = torch.randn(200, 32)
        d, i = knn(x, q, 8, res=res)
        ref_d, ref_i = torch.cdist(q, x).pow(2).topk(8, largest=False)
        torch.testing.assert_close(d, ref_d, rtol=1e-4, atol=1e-4)
        assert (i == ref_i).float().mean() > 0.99  # ties may reorder
