import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an AMD GPU (MI355X)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def device():
    return torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")


@pytest.fixture
def gpu_device():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    return torch.device("cuda")


def spawn_gloo(fn, world_size=2, args=()):
    """Run fn(rank, world_size, *args) in world_size processes over gloo.

    Used by multi-process comms/kmeans tests on CPU (the reference tests
    collectives on a real cluster only; we cover them here without GPUs).
    """
    import torch.multiprocessing as mp

    port = _free_port()
    ctx = mp.get_context("spawn")
    procs = []
    for rank in range(world_size):
        p = ctx.Process(target=_gloo_entry, args=(fn, rank, world_size, port, args))
        p.start()
        procs.append(p)
    for p in procs:
        p.join(180)
    # terminate any worker that outlived the join timeout before asserting,
    # so a hang can't leak processes past the test session
    hung = [p for p in procs if p.exitcode is None]
    for p in hung:
        p.terminate()
        p.join(10)
    for p in procs:
        assert p.exitcode == 0, f"worker exited with {p.exitcode}"


def _gloo_entry(fn, rank, world_size, port, args):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    import torch.distributed as dist

    dist.init_process_group("gloo", rank=rank, world_size=world_size)
    try:
        fn(rank, world_size, *args)
    finally:
        dist.destroy_process_group()


def _free_port():
    import socket

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port
