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


def _no_sync_worker(rank, world):
    """engine.no_sync(): grads must NOT be reduced inside the context and
    MUST be on exit (reference engine.py:2065)."""
    import deepspeed_amd
    torch.manual_seed(0)
    model = torch.nn.Linear(8, 4)
    engine, opt, _, _ = deepspeed_amd.initialize(model=model, config={
        "train_micro_batch_size_per_gpu": 2,
        "zero_optimization": {"stage": 1, "overlap_comm": False},
        "bf16": {"enabled": False},
        "optimizer": {"type": "AdamW", "params": {"lr": 1e-2}}})
    torch.manual_seed(10 + rank)  # rank-divergent data
    x = torch.randn(2, 8)
    with engine.no_sync():
        loss = engine(x).sum()
        engine.backward(loss)
    # inside no_sync nothing synced; weights identical (no step yet)
    loss2 = engine(x).sum()
    engine.backward(loss2)   # boundary: now reduces both micro-grads
    engine.step()
    import torch.distributed as td
    w = model.weight.detach().clone()
    peers = [torch.empty_like(w) for _ in range(world)]
    td.all_gather(peers, w)
    assert torch.equal(peers[0], peers[1])  # stepped with averaged grads


def test_engine_no_sync():
    run_distributed(_no_sync_worker, world_size=2)


def test_repeating_loader_and_comms_summary():
    from deepspeed_amd.runtime.dataloader import RepeatingLoader
    data = [1, 2, 3]
    it = iter(RepeatingLoader(data))
    assert [next(it) for _ in range(7)] == [1, 2, 3, 1, 2, 3, 1]

    from deepspeed_amd import comm as dist
    cl = dist.configure_comms_logger(enabled=True)
    out = cl.timed("all_reduce", 1024, 2, lambda: None)
    assert out is None
    stats = cl.summary() if hasattr(cl, "summary") else None
    dist.log_summary()  # smoke: prints without error
    dist.configure_comms_logger(enabled=False)


def _shm_worker(rank, world):
    import torch
    from deepspeed_amd.ops._loader import get_ext
    ext = get_ext()
    if ext is None or not hasattr(ext, "ShmComm"):
        return
    comm = ext.ShmComm("testgrp", rank, world, 4096)
    t = torch.full((1000,), float(rank + 1), dtype=torch.float32)
    comm.all_reduce(t)
    expect = sum(r + 1 for r in range(world))
    assert torch.all(t == expect), t[:4]
    # second call reuses the group (generation barrier correctness)
    t2 = torch.arange(100, dtype=torch.float32) * (rank + 1)
    comm.all_reduce(t2)
    assert torch.allclose(t2, torch.arange(100, dtype=torch.float32) * expect)
    del comm


def test_shm_allreduce_intranode():
    """POSIX shared-memory CPU allreduce (reference csrc/cpu/comm/shm.cpp):
    cross-process sum without touching the network stack."""
    from .common import run_distributed
    run_distributed(_shm_worker, world_size=4)


def test_mpi_discovery_env(monkeypatch):
    """OpenMPI env vars map to the torchrun-style rendezvous variables
    (reference comm/comm.py:694 mpi_discovery)."""
    import deepspeed_amd.comm as dcomm
    for k in ("RANK", "WORLD_SIZE", "LOCAL_RANK", "MASTER_ADDR",
              "MASTER_PORT"):
        monkeypatch.delenv(k, raising=False)
    monkeypatch.setenv("OMPI_COMM_WORLD_RANK", "3")
    monkeypatch.setenv("OMPI_COMM_WORLD_SIZE", "8")
    monkeypatch.setenv("OMPI_COMM_WORLD_LOCAL_RANK", "3")
    dcomm.mpi_discovery(distributed_port=12345)
    import os
    assert os.environ["RANK"] == "3"
    assert os.environ["WORLD_SIZE"] == "8"
    assert os.environ["LOCAL_RANK"] == "3"
    assert os.environ["MASTER_ADDR"] == "127.0.0.1"
    assert os.environ["MASTER_PORT"] == "12345"


def test_mpi_discovery_requires_mpi_env(monkeypatch):
    import deepspeed_amd.comm as dcomm
    for k in ("RANK", "OMPI_COMM_WORLD_RANK", "PMI_RANK"):
        monkeypatch.delenv(k, raising=False)
    import pytest as _pytest
    with _pytest.raises(RuntimeError):
        dcomm.mpi_discovery()


def test_initialize_mesh_device_ws4():
    from .common import run_distributed
    run_distributed(_mesh_worker, world_size=4)


def _mesh_worker(rank, world):
    import deepspeed_amd.comm as dcomm
    mesh = dcomm.initialize_mesh_device((2, 2))
    dp = mesh.get_group("data_parallel")
    sp = mesh.get_group("sequence_parallel")
    import torch.distributed as td
    assert td.get_world_size(dp) == 2 and td.get_world_size(sp) == 2
    # the two axes partition the 4 ranks consistently
    t = torch.tensor([float(rank)])
    td.all_reduce(t, group=sp)
    expected = {0: 1.0, 1: 1.0, 2: 5.0, 3: 5.0}[rank]
    assert t.item() == expected, (rank, t.item())
