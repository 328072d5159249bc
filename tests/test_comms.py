"""Comms verification suite — the analog of the reference's
comms/detail/test.hpp test_collective_* functions, run over gloo with 2
processes on CPU (and over RCCL on GPU via the same code path).
"""
import pytest
import torch

from raft_amd.comms import LoopbackComms, ReduceOp


class TestLoopback:
    def test_all_ops(self):
        c = LoopbackComms()
        t = torch.arange(4.0)
        assert c.get_size() == 1 and c.get_rank() == 0
        torch.testing.assert_close(c.allreduce(t.clone()), t)
        torch.testing.assert_close(c.allgather(t).squeeze(0), t)
        torch.testing.assert_close(c.reducescatter(t), t)
        assert c.comm_split(0, 0) is c


def _collective_worker(rank, world):
    from raft_amd.comms import TorchDistComms
    from tests.comms_suite import collective_suite

    c = TorchDistComms()
    assert c.get_size() == world
    assert c.get_rank() == rank
    collective_suite(c, "cpu")


def test_collectives_gloo_world2():
    from tests.conftest import spawn_gloo

    spawn_gloo(_collective_worker, world_size=2)


def _kmeans_worker(rank, world):
    from raft_amd.comms import TorchDistComms
    from raft_amd.cluster import kmeans_fit, KMeansParams
    from raft_amd.random import make_blobs, RngState

    torch.manual_seed(0)
    # identical global dataset on all ranks; each takes its shard
    x, y, centers = make_blobs(600, 5, n_clusters=4, cluster_std=0.3,
                               center_box=(-10, 10), state=RngState(seed=11))
    shard = x[rank * 300:(rank + 1) * 300]
    comms = TorchDistComms()
    model = kmeans_fit(shard, KMeansParams(n_clusters=4, max_iter=30, seed=3,
                                           init="kmeans++"), comms=comms)
    # distributed fit must reach the true centers
    d = torch.cdist(centers, model.centroids)
    assert d.min(dim=1).values.max() < 1.0, d.min(dim=1).values
    # all ranks end with identical centroids
    g = comms.allgather(model.centroids)
    assert torch.allclose(g[0], g[world - 1], atol=1e-6)


def test_distributed_kmeans_gloo_world2():
    from tests.conftest import spawn_gloo

    spawn_gloo(_kmeans_worker, world_size=2)


def _scalable_init_worker(rank, world):
    from raft_amd.comms import TorchDistComms
    from raft_amd.cluster import kmeans_fit, KMeansParams
    from raft_amd.random import make_blobs, RngState

    x, _, centers = make_blobs(800, 6, n_clusters=5, cluster_std=0.3,
                               center_box=(-12, 12), state=RngState(seed=21))
    shard = x[rank * 400:(rank + 1) * 400]
    comms = TorchDistComms()
    model = kmeans_fit(shard, KMeansParams(n_clusters=5, max_iter=25, seed=2,
                                           init="scalable"), comms=comms)
    d = torch.cdist(centers, model.centroids)
    assert d.min(dim=1).values.max() < 1.0, d.min(dim=1).values
    g = comms.allgather(model.centroids)
    assert torch.allclose(g[0], g[world - 1], atol=1e-6)


def test_distributed_scalable_init_gloo_world2():
    from tests.conftest import spawn_gloo

    spawn_gloo(_scalable_init_worker, world_size=2)


class TestLoopbackFidelity:
    """LoopbackComms must be shape/dtype/value-faithful for every op the
    algorithms use — it stands in for RCCL in most unit tests."""

    def test_all_ops_shapes_dtypes(self):
        from raft_amd.comms import LoopbackComms, ReduceOp
        c = LoopbackComms()
        assert c.get_size() == 1 and c.get_rank() == 0
        for dtype in (torch.float32, torch.float64, torch.int64):
            for shape in ((3,), (2, 5), (1,)):
                t = (torch.randn(shape) * 10).to(dtype)
                orig = t.clone()
                c.allreduce(t, op=ReduceOp.SUM)
                assert torch.equal(t, orig)          # world=1: identity
                g = c.allgather(orig)
                assert g.shape == (1,) + tuple(shape)
                assert torch.equal(g[0], orig)
                b = c.bcast(orig.clone(), root=0)
                assert torch.equal(b, orig)
        v = torch.arange(6, dtype=torch.float32).reshape(3, 2)
        av = c.allgatherv(v, [3])
        assert torch.equal(av, v)
        c.barrier()
        sub = c.comm_split(color=0, key=0)
        assert sub.get_size() == 1


def test_interface_fully_overridden():
    """Every abstract comms_t op must be overridden by both concrete
    communicators (reference comms_iface completeness)."""
    import inspect
    from raft_amd.comms.comms import Comms, LoopbackComms, TorchDistComms
    abstract = [n for n, f in vars(Comms).items()
                if callable(f) and "NotImplementedError" in inspect.getsource(f)]
    assert len(abstract) >= 15
    for impl in (LoopbackComms, TorchDistComms):
        missing = [n for n in abstract if getattr(impl, n) is getattr(Comms, n)]
        assert not missing, f"{impl.__name__} missing {missing}"


def test_collectives_gloo_world3():
    """Odd world size: ragged allgatherv/gatherv chunks and the even/odd
    comm_split take different branches than world=2."""
    from tests.conftest import spawn_gloo

    spawn_gloo(_collective_worker, world_size=3)


def _mpi_style_entry(rank, world, port):
    """Simulate an mpirun-launched process: only OMPI_* vars set."""
    import os
    for k in ("RANK", "WORLD_SIZE", "LOCAL_RANK"):
        os.environ.pop(k, None)
    os.environ["OMPI_COMM_WORLD_RANK"] = str(rank)
    os.environ["OMPI_COMM_WORLD_SIZE"] = str(world)
    os.environ["OMPI_COMM_WORLD_LOCAL_RANK"] = str(rank)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    from raft_amd.comms import init_comms
    c = init_comms(backend="gloo")
    assert c.get_size() == world and c.get_rank() == rank
    t = torch.ones(3)
    c.allreduce(t)
    assert torch.equal(t, torch.full((3,), float(world)))
    import torch.distributed as dist
    dist.destroy_process_group()


def test_mpi_launcher_bootstrap():
    """mpi_comms parity (reference comms/mpi_comms.hpp:50): processes with
    only MPI rank env vars rendezvous through the same init()."""
    import torch.multiprocessing as mp
    from tests.conftest import _free_port

    port = _free_port()
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_mpi_style_entry, args=(r, 2, port))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(180)
    for p in procs:
        assert p.exitcode == 0, f"worker exited with {p.exitcode}"
