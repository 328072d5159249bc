"""Device-generic comms_t verification suite.

The analog of the reference's comms/detail/test.hpp test_collective_* /
test_pointToPoint_* functions (one assertion block per op). Runs over gloo
with CPU tensors (CI) and over RCCL with device tensors (tests/test_gpu_comms.py
launches it under torchrun on an MI355X).
"""
import torch


def collective_suite(c, device):
    """Run every comms_t op with tensors on `device` and assert results."""
    from raft_amd.comms import ReduceOp

    world = c.get_size()
    rank = c.get_rank()
    dev = torch.device(device)

    # allreduce (test_collective_allreduce parity)
    t = torch.full((4,), float(rank + 1), device=dev)
    c.allreduce(t)
    expected = sum(r + 1 for r in range(world))
    assert torch.equal(t, torch.full((4,), float(expected), device=dev))

    # async allreduce (the kmeans split-batch overlap path): two outstanding
    # collectives issued back to back, waited in order
    ta = torch.full((8,), float(rank + 1), device=dev)
    tb = torch.full((8,), 2.0 * (rank + 1), device=dev)
    wa = c.allreduce_async(ta)
    wb = c.allreduce_async(tb)
    if wa is not None:
        wa.wait()
    if wb is not None:
        wb.wait()
    assert torch.equal(ta, torch.full((8,), float(expected), device=dev))
    assert torch.equal(tb, torch.full((8,), 2.0 * expected, device=dev))

    # allreduce MAX (the bench elapsed-time reduction path)
    t = torch.full((1,), float(rank), device=dev, dtype=torch.float64)
    c.allreduce(t, op=ReduceOp.MAX)
    assert torch.equal(t, torch.full((1,), float(world - 1), device=dev,
                                     dtype=torch.float64))

    # bcast
    t = torch.full((3,), float(rank), device=dev)
    c.bcast(t, root=0)
    assert torch.equal(t, torch.zeros(3, device=dev))

    # reduce
    t = torch.full((2,), 1.0, device=dev)
    c.reduce(t, root=0, op=ReduceOp.SUM)
    if rank == 0:
        assert torch.equal(t, torch.full((2,), float(world), device=dev))

    # allgather
    g = c.allgather(torch.full((2,), float(rank), device=dev))
    for r in range(world):
        assert torch.equal(g[r], torch.full((2,), float(r), device=dev))

    # allgatherv (ragged -> batched P2P path on nccl/rccl)
    counts = [r + 1 for r in range(world)]
    mine = torch.full((rank + 1,), float(rank), device=dev)
    cat = c.allgatherv(mine, counts)
    assert cat.numel() == sum(counts)
    off = 0
    for r in range(world):
        assert torch.equal(cat[off:off + r + 1],
                           torch.full((r + 1,), float(r), device=dev))
        off += r + 1

    # gather / gatherv
    got = c.gather(torch.full((2,), float(rank), device=dev), root=0)
    if rank == 0:
        assert got.shape[0] == world
    gv = c.gatherv(mine, counts, root=0)
    if rank == 0:
        assert gv.numel() == sum(counts)

    # reducescatter
    t = torch.arange(float(world * 2), device=dev)
    out = c.reducescatter(t.clone())
    assert torch.equal(out, t[rank * 2:(rank + 1) * 2] * world)

    if world > 1:
        # p2p sendrecv ring (self-send at world 1 is undefined for NCCL/RCCL)
        send = torch.full((3,), float(rank), device=dev)
        recv = torch.empty(3, device=dev)
        dst = (rank + 1) % world
        src = (rank - 1) % world
        c.device_sendrecv(send, dst, recv, src)
        assert torch.equal(recv, torch.full((3,), float(src), device=dev))

        # multicast sendrecv: each rank sends to every peer, receives from all
        msend = torch.full((2,), float(rank) + 10.0, device=dev)
        peers = [r for r in range(world) if r != rank]
        mrecvs = [torch.empty(2, device=dev) for _ in peers]
        c.device_multicast_sendrecv(msend, peers, mrecvs, peers)
        for r, peer in zip(mrecvs, peers):
            assert torch.equal(r, torch.full((2,), float(peer) + 10.0,
                                             device=dev))

    # comm_split: even/odd colors
    sub = c.comm_split(color=rank % 2, key=rank)
    t = torch.ones(1, device=dev)
    sub.allreduce(t)
    n_same_color = len([r for r in range(world) if r % 2 == rank % 2])
    assert torch.equal(t, torch.full((1,), float(n_same_color), device=dev))

    c.barrier()
