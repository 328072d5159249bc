import io

import numpy as np
import pytest
import torch

from raft_amd.core import (
    Resources, DeviceResources, Handle, DeviceResourcesManager, get_resources,
    device_ndarray, Bitset, serialize_mdspan, deserialize_mdspan,
)
from raft_amd.core.serialize import dumps, loads
from raft_amd.core.interruptible import Interruptible, InterruptedException


class TestResources:
    def test_lazy_registry(self):
        res = Resources(device=torch.device("cpu"))
        calls = []
        res.add_resource_factory("thing", lambda: calls.append(1) or "made")
        assert not calls
        assert res.get_resource("thing") == "made"
        assert res.get_resource("thing") == "made"
        assert len(calls) == 1  # lazily created exactly once

    def test_clone_respecializes(self):
        res = Resources(device=torch.device("cpu"))
        res.add_resource_factory("x", lambda: object())
        a = res.get_resource("x")
        c = res.clone()
        assert c.get_resource("x") is not a  # fresh instance in the clone

    def test_handle_alias(self):
        assert Handle is DeviceResources

    def test_manager_round_robin(self):
        mgr = DeviceResourcesManager(pool_size=2)
        r1 = mgr.get_resources(torch.device("cpu"))
        r2 = mgr.get_resources(torch.device("cpu"))
        r3 = mgr.get_resources(torch.device("cpu"))
        assert r1 is not r2
        assert r3 is r1

    def test_get_resources_default(self):
        r = get_resources()
        assert isinstance(r, Resources)
        assert get_resources(r) is r

    def test_comms_injection(self):
        from raft_amd.comms import LoopbackComms, inject_comms
        res = Resources(device=torch.device("cpu"))
        assert not res.has_comms()
        with pytest.raises(RuntimeError):
            res.get_comms()
        inject_comms(res, LoopbackComms())
        assert res.get_comms().get_size() == 1


class TestDeviceNdarray:
    def test_roundtrip_numpy(self):
        a = np.random.rand(4, 5).astype(np.float32)
        d = device_ndarray(a)
        assert d.shape == (4, 5)
        assert d.dtype == np.float32
        np.testing.assert_array_equal(d.copy_to_host(), a)

    def test_empty_zeros(self):
        z = device_ndarray.zeros((3, 2), dtype=np.float64, device="cpu")
        assert z.copy_to_host().sum() == 0.0
        e = device_ndarray.empty((2, 2), device="cpu")
        assert e.shape == (2, 2)

    def test_dlpack(self):
        a = np.arange(6, dtype=np.int32).reshape(2, 3)
        d = device_ndarray(a)
        t = torch.from_dlpack(d)
        np.testing.assert_array_equal(t.numpy(), a)


class TestSerialize:
    def test_npy_roundtrip_matches_numpy(self):
        t = torch.randn(5, 7, dtype=torch.float64)
        data = dumps(t)
        # cross-check: numpy can read our bytes
        arr = np.load(io.BytesIO(data))
        np.testing.assert_array_equal(arr, t.numpy())
        t2 = loads(data)
        assert torch.equal(t, t2)

    def test_file_roundtrip(self, tmp_path):
        from raft_amd.core import save_npy, load_npy
        t = torch.randint(0, 100, (8,), dtype=torch.int64)
        p = str(tmp_path / "x.npy")
        save_npy(p, t)
        assert torch.equal(load_npy(p), t)


class TestBitset:
    def test_set_test_count(self):
        bs = Bitset(100, default=False)
        idx = torch.tensor([0, 5, 31, 32, 63, 64, 99])
        bs.set(idx)
        assert bs.test(idx).all()
        assert bs.count() == 7
        bs.set(torch.tensor([5]), value=False)
        assert bs.count() == 6

    def test_flip_and_tail_mask(self):
        bs = Bitset(33, default=False)
        bs.flip()
        assert bs.count() == 33  # tail bits beyond n must not count

    def test_dense_roundtrip(self):
        dense = torch.rand(77) > 0.5
        bs = Bitset.from_dense(dense)
        assert torch.equal(bs.to_dense(), dense)
        assert bs.count() == int(dense.sum())
        assert abs(bs.sparsity() - (1 - dense.float().mean().item())) < 1e-6


class TestInterruptible:
    def test_cancel_raises(self):
        tok = Interruptible()
        tok.cancel()
        with pytest.raises(InterruptedException):
            tok.check()
        tok.check()  # flag cleared after raise

    def test_sync_noop_on_cpu(self):
        Interruptible().synchronize()

    def test_cross_thread_cancel(self):
        """reference semantics: cancel() from ANOTHER thread interrupts the
        spinning waiter (interruptible.hpp:64 token-per-thread design)."""
        import threading
        import time as _t
        tok = Interruptible()
        hit = {}

        def waiter():
            try:
                for _ in range(20000):
                    tok.check()
                    _t.sleep(0.001)
                hit["r"] = "timeout"
            except InterruptedException:
                hit["r"] = "interrupted"

        th = threading.Thread(target=waiter)
        th.start()
        _t.sleep(0.05)
        tok.cancel()
        th.join(30)
        assert hit.get("r") == "interrupted"


class TestCompat:
    def test_pylibraft_style_imports(self):
        from raft_amd import compat
        import scipy.sparse as sp
        import numpy as np
        rng = np.random.RandomState(0)
        s = sp.random(80, 80, density=0.1, random_state=rng, format="csr",
                      dtype=np.float64)
        s = (s + s.T) * 0.5 + sp.eye(80) * 0.2
        w, v = compat.eigsh(s.tocsr(), k=3, tol=1e-9)
        import scipy.sparse.linalg as spla
        ref = np.sort(spla.eigsh(s.tocsr(), k=3, which="SA")[0])
        np.testing.assert_allclose(np.asarray(w), ref, rtol=1e-4, atol=1e-6)

    def test_compat_pairwise_and_select(self):
        from raft_amd import compat
        import numpy as np
        x = np.random.rand(10, 4).astype(np.float32)
        d = compat.pairwise_distance(x, x, metric="sqeuclidean")
        assert np.asarray(d).shape == (10, 10)
        v, i = compat.select_k(np.asarray(d), 3)
        assert np.asarray(v).shape == (10, 3)

    def test_compat_rmat(self):
        from raft_amd import compat
        import numpy as np
        out = np.zeros((1000, 2), dtype=np.int64)
        theta = np.array([0.6, 0.2, 0.15, 0.05] * 8, dtype=np.float64)
        compat.rmat(out, theta, 8, 8, seed=7)
        assert out.max() < 256 and out.min() >= 0


class TestTraceMemoryMdbuffer:
    def test_annotate_noop_on_cpu(self):
        from raft_amd.core import annotate, annotated

        with annotate("x"):
            pass

        @annotated("y")
        def f(a):
            return a + 1

        assert f(1) == 2

    def test_memory_stats_and_monitor(self):
        from raft_amd.core import MemoryStats, TrackingScope, ResourceMonitor
        import time
        s = MemoryStats.capture()
        assert s.allocated_bytes >= 0
        with TrackingScope() as ts:
            _ = torch.zeros(10)
        assert ts.delta_allocated >= 0
        with ResourceMonitor(period_s=0.01) as mon:
            time.sleep(0.05)
        assert len(mon.samples) >= 2

    def test_mdbuffer_lazy_views(self):
        from raft_amd.core import MDBuffer, MemoryType, memory_type_dispatcher
        t = torch.arange(6, dtype=torch.float32)
        buf = MDBuffer(t)
        assert buf.memory_type == MemoryType.HOST
        h = buf.view(MemoryType.HOST)
        assert h is t
        out = memory_type_dispatcher(buf, lambda x: "dev", lambda x: "host")
        assert out == "host"

    def test_print_matrix(self):
        from raft_amd.matrix import print_matrix
        s = print_matrix(torch.eye(3), name="I")
        assert "I" in s and "1" in s


class TestOperatorsErrorKvpMath:
    def test_operators_compose(self):
        from raft_amd.core import operators as op
        x = torch.tensor([1.0, -4.0, 9.0])
        assert torch.equal(op.abs_op(x), x.abs())
        assert torch.equal(op.sq_op(x), x * x)
        f = op.compose_op(op.sqrt_op, op.abs_op)   # sqrt(abs(x))
        torch.testing.assert_close(f(x), x.abs().sqrt())
        add3 = op.add_const_op(3.0)
        torch.testing.assert_close(add3(x), x + 3)
        sqd = op.sqdiff_op
        torch.testing.assert_close(sqd(x, torch.ones(3)), (x - 1) ** 2)
        assert op.div_checkzero_op(1.0, 0.0) == 0.0
        t = op.div_checkzero_op(torch.ones(2), torch.tensor([0.0, 2.0]))
        assert torch.equal(t, torch.tensor([0.0, 0.5]))

    def test_kvp_argmin_op(self):
        from raft_amd.core import KeyValuePair
        from raft_amd.core.operators import argmin_op, argmax_op
        a, b = KeyValuePair(3, 1.5), KeyValuePair(1, 1.5)
        assert argmin_op(a, b).key == 1        # tie -> smaller key
        assert argmax_op(a, KeyValuePair(0, 2.0)).value == 2.0
        k, v = KeyValuePair("k", 7)
        assert (k, v) == ("k", 7)

    def test_error_expects(self):
        from raft_amd.core import expects, fail, LogicError, RaftError
        expects(True)
        with pytest.raises(LogicError):
            expects(False, "nope")
        with pytest.raises(RaftError):
            fail("boom")

    def test_math_wrappers(self):
        from raft_amd.core import math as rmath
        assert rmath.sigmoid(0.0) == pytest.approx(0.5)
        assert rmath.sigmoid(-800.0) == pytest.approx(0.0)  # no overflow
        x = torch.tensor([0.0, 1.0])
        torch.testing.assert_close(rmath.sigmoid(x), torch.sigmoid(x))
        assert rmath.max(2, 5) == 5 and rmath.min(2, 5) == 2
        torch.testing.assert_close(rmath.atan2(torch.ones(2), torch.ones(2)),
                                   torch.full((2,), 0.7853981633974483))

    def test_temporary_device_buffer(self):
        from raft_amd.core import TemporaryDeviceBuffer
        x = torch.arange(4, dtype=torch.float32)
        with TemporaryDeviceBuffer(x, device="cpu", write_back=True) as buf:
            v = buf.view()
            v += 1
        assert torch.equal(x, torch.tensor([1.0, 2.0, 3.0, 4.0]))


class TestWorkspaceResource:
    def test_workspace_accessors(self):
        from raft_amd.core import Resources
        r = Resources(torch.device("cpu"))
        with r.get_workspace((16, 4), dtype=torch.float32) as w:
            t = w.view((16, 4), torch.float32)
            assert t.shape == (16, 4) and t.dtype == torch.float32
        assert r.workspace_stats() == (0, 0)   # cpu: no device pool
        r.empty_workspace_pool()               # no-op, must not raise


class TestCompatCommon:
    def test_stream_handle_output_as(self):
        from raft_amd import compat
        s = compat.Stream()
        s.sync()   # cpu no-op
        calls = {}

        @compat.auto_sync_handle
        def takes_handle(x, handle=None):
            calls["handle"] = handle
            return x * 2

        out = takes_handle(torch.ones(2))
        assert calls["handle"] is not None
        assert torch.equal(out, torch.full((2,), 2.0))
        compat.set_output_as("array")
        import numpy as np
        assert isinstance(compat.post_output(torch.ones(2)), np.ndarray)
        compat.set_output_as("torch")
        assert torch.is_tensor(compat.post_output(torch.ones(2)))
        assert float(compat.interruptible(lambda: torch.ones(1).sum())) == 1.0


class TestUtils:
    def test_helpers(self):
        from raft_amd.utils import ceil_div, next_pow2, row_chunks, as_2d
        assert ceil_div(10, 3) == 4 and ceil_div(9, 3) == 3
        assert next_pow2(1) == 1 and next_pow2(5) == 8 and next_pow2(64) == 64
        chunks = list(row_chunks(10, 4))
        assert chunks == [(0, 4), (4, 8), (8, 10)]
        assert as_2d(torch.arange(6)).shape[0] == 1 or as_2d(torch.arange(6)).dim() == 2


class TestCopyMdspan:
    def test_layout_dtype_conversion(self):
        from raft_amd.core import copy_mdspan
        src = torch.arange(12, dtype=torch.float64).reshape(3, 4)
        # layout conversion: dst is a transposed (non-contiguous) view target
        dst = torch.empty(4, 3, dtype=torch.float32).t()
        copy_mdspan(dst, src)
        torch.testing.assert_close(dst, src.float())
        assert not dst.is_contiguous()
        with pytest.raises(ValueError):
            copy_mdspan(torch.empty(2, 2), src)


class TestSpan:
    def test_spans(self):
        from raft_amd.core import host_span, subspan, device_span
        x = torch.arange(12, dtype=torch.float32).reshape(3, 4)
        s = host_span(x)
        assert s.data_ptr() == x.data_ptr() and s.numel() == 12
        sub = subspan(s, 4, 4)
        assert torch.equal(sub, torch.tensor([4.0, 5, 6, 7]))
        sub[0] = 99.0                      # non-owning: writes through
        assert x[1, 0] == 99.0
        with pytest.raises(IndexError):
            subspan(s, 10, 5)
        with pytest.raises(TypeError):
            device_span(x)                 # host tensor -> must raise on CPU


class TestResourcesStreamPool:
    def test_pool_round_robin_indexing(self):
        from raft_amd.core import Resources
        r = Resources(torch.device("cpu"), stream_pool_size=0)
        # empty pool: stream_from_pool falls back to the main stream (None on CPU)
        assert r.stream_from_pool(3) is r.stream
        # type-indexed registry: custom resources are lazy and cached
        calls = []
        r.add_resource_factory("my_res", lambda: calls.append(1) or {"x": 1})
        a = r.get_resource("my_res")
        b = r.get_resource("my_res")
        assert a is b and calls == [1]

    def test_clone_shares_factories(self):
        from raft_amd.core import Resources
        r = Resources(torch.device("cpu"))
        r.add_resource_factory("shared", lambda: object())
        c = r.clone()
        assert c.get_resource("shared") is not None


class TestLoggerEnvSinks:
    def test_file_sink_and_level_env(self, tmp_path):
        """RAFT_AMD_DEBUG file sink + RAFT_AMD_LOG_LEVEL (reference RAFT_DEBUG
        env + compile-time level, logger.hpp:25-49) — in a subprocess so the
        module-level logger singleton initializes from the env."""
        import subprocess
        import sys as _sys
        f = tmp_path / "raft.log"
        code = (
            "from raft_amd.core import get_logger\n"
            "log = get_logger()\n"
            "log.info('should-appear')\n"
            "log.debug('should-not-appear')\n"
        )
        r = subprocess.run([_sys.executable, "-c", code], env={
            **__import__("os").environ,
            "RAFT_AMD_DEBUG": str(f), "RAFT_AMD_LOG_LEVEL": "info"},
            capture_output=True, timeout=120)
        assert r.returncode == 0, r.stderr.decode()
        text = f.read_text()
        assert "should-appear" in text and "should-not-appear" not in text


class TestManagerRoundRobin:
    def test_pool_bounded_and_cycling(self):
        mgr = DeviceResourcesManager(pool_size=3)
        seen = [mgr.get_resources(torch.device("cpu")) for _ in range(9)]
        distinct = {id(r) for r in seen}
        assert len(distinct) == 3            # pool capped at pool_size
        # round-robin: i and i+3 land on the same Resources
        for i in range(6):
            assert seen[i] is seen[i + 3]

    def test_singleton_instance(self):
        a = DeviceResourcesManager.instance()
        b = DeviceResourcesManager.instance()
        assert a is b


class TestDeviceNdarraySurface:
    def test_constructors_and_props(self):
        import numpy as np
        a = device_ndarray.zeros((3, 4), dtype=np.float32)
        assert a.shape == (3, 4) and a.c_contiguous
        b = device_ndarray.empty((2, 2), dtype=np.float64)
        assert b.torch.dtype == torch.float64
        c = device_ndarray(np.arange(6, dtype=np.int32).reshape(2, 3))
        assert torch.equal(c.torch, torch.arange(6, dtype=torch.int32).reshape(2, 3))
        # numpy round trip preserves values and dtype
        back = np.asarray(c.copy_to_host() if hasattr(c, "copy_to_host")
                          else c.torch.cpu().numpy())
        assert back.dtype == np.int32 and back.sum() == 15
