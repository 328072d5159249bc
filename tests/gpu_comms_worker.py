"""torchrun-launched RCCL verification worker (one rank per process).

Launched by tests/test_gpu_comms.py as
    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        --master-addr 127.0.0.1 tests/gpu_comms_worker.py
On a 1-GPU box all ranks share device 0 (multiple ranks per device); on an
8-GPU node each rank pins its LOCAL_RANK device. This executes the REAL RCCL
("nccl") branch of TorchDistComms — every collective the algorithms use —
plus the full distributed k-means iteration (packed allreduce, bcast), i.e.
exactly what bench.py --gpus N runs.

Reference parity: comms/detail/test.hpp:31-529 perform_test_comms_* executed
on real NCCL via raft-dask's LocalCUDACluster (conftest.py:15-35).
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    assert torch.cuda.is_available(), "RCCL worker needs a GPU"
    dev_index = local_rank % max(torch.cuda.device_count(), 1)
    torch.cuda.set_device(dev_index)
    dev = torch.device("cuda", dev_index)

    from raft_amd.comms import init_comms
    from tests.comms_suite import collective_suite

    c = init_comms()  # backend "nccl" == RCCL on ROCm
    assert c.get_size() == world and c.get_rank() == rank
    import torch.distributed as dist
    assert dist.get_backend() == "nccl", dist.get_backend()

    collective_suite(c, dev)
    torch.cuda.synchronize()
    print(f"[rank {rank}] RCCL collective suite OK", flush=True)

    # flagship-shaped packed allreduce ([k*d + k + 1] fp32 — the kmeans
    # per-iteration collective) through RCCL explicitly, so even a world-1
    # run launches the real ncclAllReduce on a production-sized buffer
    k_, d_ = 1024, 256
    g = torch.Generator(device="cpu").manual_seed(1234)   # identical per rank
    packed = torch.randn(k_ * d_ + k_ + 1, generator=g).to(dev)
    ref = packed.clone()
    c.allreduce(packed)
    torch.cuda.synchronize()
    assert torch.allclose(packed, ref * world, rtol=1e-5, atol=1e-5)

    # ---- distributed k-means on device: the bench.py --gpus N inner loop ----
    # force the split-batch allreduce overlap so the async path is exercised
    # on multi-GPU nodes regardless of shard size
    os.environ["RAFT_AMD_KMEANS_OVERLAP"] = "1"
    from raft_amd.cluster.kmeans import kmeans_iterate
    from raft_amd.comms import LoopbackComms
    from raft_amd.random import make_blobs, RngState

    n_global, d, k = 65536, 64, 128
    x, _, centers = make_blobs(n_global, d, n_clusters=k, cluster_std=0.5,
                               state=RngState(seed=7), device=dev)
    shard = n_global // world
    x_local = x[rank * shard:(rank + 1) * shard].contiguous()
    c0 = (centers + 0.25).contiguous()

    cd, inertia_d = kmeans_iterate(x_local, c0.clone(), 3, comms=c,
                                   fp32_mode="bf16x2v")
    torch.cuda.synchronize()

    # all ranks must hold identical centroids (same allreduced update)
    g = c.allgather(cd)
    assert torch.equal(g[0], g[world - 1]), "ranks diverged"

    # the distributed result must match the single-process run on the full
    # data (identical fp32 sums, allreduce only changes the summation split)
    cs, inertia_s = kmeans_iterate(x, c0.clone(), 3, comms=LoopbackComms(),
                                   fp32_mode="bf16x2v")
    rel = abs(inertia_d - inertia_s) / max(abs(inertia_s), 1e-30)
    assert rel < 1e-4, (inertia_d, inertia_s)
    assert torch.allclose(cd, cs, rtol=1e-4, atol=1e-4)
    print(f"[rank {rank}] distributed kmeans OK "
          f"(inertia {inertia_d:.6e} vs single {inertia_s:.6e})", flush=True)

    # the adaptive auto engine under real comms: per-rank engine decisions
    # are independent but every verified engine is exact, so the distributed
    # result must still match the single-process run
    ca, inertia_a = kmeans_iterate(x_local, c0.clone(), 3, comms=c,
                                   fp32_mode="auto")
    torch.cuda.synchronize()
    rel_a = abs(inertia_a - inertia_s) / max(abs(inertia_s), 1e-30)
    assert rel_a < 1e-4, (inertia_a, inertia_s)
    assert torch.allclose(ca, cs, rtol=1e-4, atol=1e-4)
    print(f"[rank {rank}] distributed kmeans (auto engine) OK", flush=True)

    # mixed setup: a gloo subgroup on a GPU node with CUDA tensors must
    # host-stage instead of faulting (control-plane traffic pattern)
    import torch.distributed as dist
    gloo_group = dist.new_group(backend="gloo")
    from raft_amd.comms.comms import TorchDistComms
    gc = TorchDistComms(gloo_group)
    t = torch.full((64,), float(rank + 1), device=dev)
    gc.allreduce(t)
    expect = sum(range(1, world + 1))
    assert torch.allclose(t, torch.full_like(t, float(expect)))
    g2 = gc.allgather(torch.tensor([float(rank)], device=dev))
    assert g2.shape == (world, 1) and float(g2[rank, 0]) == float(rank)
    b = torch.full((8,), float(rank), device=dev)
    gc.bcast(b, root=0)
    assert torch.allclose(b, torch.zeros_like(b))
    print(f"[rank {rank}] gloo-on-GPU host-staged collectives OK", flush=True)

    dist.barrier()
    dist.destroy_process_group()
    print(f"[rank {rank}] GPU_COMMS_WORKER_OK", flush=True)


if __name__ == "__main__":
    main()
