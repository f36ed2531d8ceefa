"""Comm facade tests on gloo world_size=2 (reference: tests/unit/comm/)."""

import torch

from .common import run_distributed


def _allreduce_worker(rank, world):
    import deepspeed_amd.comm as dist
    t = torch.ones(17) * (rank + 1)
    dist.all_reduce(t)
    assert torch.allclose(t, torch.full((17,), 3.0))


def _broadcast_worker(rank, world):
    import deepspeed_amd.comm as dist
    t = torch.arange(8.0) if rank == 0 else torch.zeros(8)
    dist.broadcast(t, src=0)
    assert torch.allclose(t, torch.arange(8.0))


def _reduce_scatter_worker(rank, world):
    import deepspeed_amd.comm as dist
    full = torch.arange(8.0) + rank  # rank r: [r, r+1, ..., r+7]
    out = torch.zeros(4)
    dist.reduce_scatter_tensor(out, full)
    # sum over ranks = 2*arange + 1, rank r owns elements [4r, 4r+4)
    expect = 2 * torch.arange(8.0) + 1
    assert torch.allclose(out, expect[rank * 4:(rank + 1) * 4])


def _allgather_worker(rank, world):
    import deepspeed_amd.comm as dist
    shard = torch.full((3,), float(rank))
    out = torch.zeros(6)
    dist.all_gather_into_tensor(out, shard)
    assert torch.allclose(out, torch.tensor([0., 0., 0., 1., 1., 1.]))


def _alltoall_worker(rank, world):
    import deepspeed_amd.comm as dist
    inp = torch.arange(4.0) + 10 * rank  # rank r sends [10r..10r+1], [10r+2..]
    out = torch.zeros(4)
    dist.all_to_all_single(out, inp)
    if rank == 0:
        assert torch.allclose(out, torch.tensor([0., 1., 10., 11.]))
    else:
        assert torch.allclose(out, torch.tensor([2., 3., 12., 13.]))


def test_all_reduce_ws2():
    run_distributed(_allreduce_worker, world_size=2)


def test_broadcast_ws2():
    run_distributed(_broadcast_worker, world_size=2)


def test_reduce_scatter_ws2():
    run_distributed(_reduce_scatter_worker, world_size=2)


def test_all_gather_ws2():
    run_distributed(_allgather_worker, world_size=2)


def test_all_to_all_ws2():
    run_distributed(_alltoall_worker, world_size=2)


def test_single_process_noop_paths():
    import deepspeed_amd.comm as dist
    # without init, collectives are no-ops / local copies
    t = torch.ones(4)
    dist.all_reduce(t)
    out = torch.zeros(4)
    dist.all_gather_into_tensor(out, torch.ones(4))
    assert torch.allclose(out, torch.ones(4))
    assert dist.get_rank() == 0 and dist.get_world_size() == 1


def _rs_coalesced_worker(rank, world):
    from deepspeed_amd import comm as dist
    torch.manual_seed(0)  # same tensors everywhere
    a = torch.arange(10, dtype=torch.float32) + rank   # rank-divergent
    b = torch.arange(7, dtype=torch.float32) * (rank + 1)
    parts = dist.reduce_scatter_coalesced([a, b])
    # expected: sum over ranks, my slice (ceil sizes, zero-padded)
    a_sum = torch.stack([torch.arange(10, dtype=torch.float32) + r
                         for r in range(world)]).sum(0)
    b_sum = torch.stack([torch.arange(7, dtype=torch.float32) * (r + 1)
                         for r in range(world)]).sum(0)
    pa, pb = 5, 4
    want_a = a_sum[rank * pa:(rank + 1) * pa]
    b_pad = torch.cat([b_sum, torch.zeros(world * pb - 7)])
    want_b = b_pad[rank * pb:(rank + 1) * pb]
    torch.testing.assert_close(parts[0], want_a)
    torch.testing.assert_close(parts[1], want_b)


def test_reduce_scatter_coalesced():
    run_distributed(_rs_coalesced_worker, world_size=2)
