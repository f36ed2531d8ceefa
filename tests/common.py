"""Fork-per-test distributed harness (parity with the reference's
tests/unit/common.py DistributedExec/DistributedTest, re-implemented on
torch.multiprocessing.spawn + gloo/RCCL with 127.0.0.1 rendezvous)."""

import os
import socket

import torch
import torch.multiprocessing as mp


def _free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _worker(rank, world_size, port, backend, fn, args):
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import deepspeed_amd.comm as dist
    from deepspeed_amd.parallel import groups
    groups.reset_groups()
    dist.init_distributed(dist_backend=backend, set_device=(backend == "nccl"))
    try:
        fn(rank, world_size, *args)
    finally:
        dist.barrier()
        dist.destroy_process_group()


def run_distributed(fn, world_size=2, backend=None, args=(), timeout=300):
    """Spawn `world_size` processes, each running fn(rank, world_size, *args).

    Joins with a deadline: a hung rank (lost collective, deadlocked
    barrier) kills the whole group and fails the test instead of hanging
    the suite forever (the reference's DistributedExec hang detection,
    tests/unit/common.py:170).
    """
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    port = _free_port()
    # fork is much faster on the CPU test host; GPU requires spawn
    method = "spawn" if torch.cuda.is_available() else "fork"
    ctx = mp.start_processes(_worker,
                             args=(world_size, port, backend, fn, args),
                             nprocs=world_size, join=False,
                             start_method=method)
    import time
    deadline = time.monotonic() + timeout
    while not ctx.join(timeout=max(1.0, deadline - time.monotonic())):
        if time.monotonic() >= deadline:
            for p in ctx.processes:
                if p.is_alive():
                    p.terminate()
            for p in ctx.processes:
                p.join(5)
            raise RuntimeError(
                f"distributed test hung: {world_size} ranks did not finish "
                f"within {timeout}s (backend={backend})")


def run_local(fn, backend=None, args=()):
    """Single-worker distributed run (world_size=1).

    Runs in a child process like run_distributed: initializing gloo in the
    pytest main process would make later fork-based multi-process tests
    deadlock (gloo background threads do not survive fork, but their locked
    mutexes do)."""
    run_distributed(fn, world_size=1, backend=backend, args=args)
