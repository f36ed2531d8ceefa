import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

# The distributed harness (tests/common.py) forks workers for speed on the
# CPU host. A fork from a process whose OpenMP pool has already spawned
# threads deadlocks the child inside the first torch op (the pool's mutexes
# are copied locked). Keep the pytest parent single-threaded so tests may
# freely run torch ops in-process AND fork afterwards.
os.environ.setdefault("OMP_NUM_THREADS", "1")

# Tests that initialize the engine IN the pytest process rendezvous on
# MASTER_PORT (default 29500). Pick a random free port once per pytest
# session so two concurrent pytest runs (or a stray daemon on 29500)
# cannot cross-connect to each other's TCPStore and deadlock.
if "MASTER_PORT" not in os.environ:
    import socket
    with socket.socket() as _s:
        _s.bind(("127.0.0.1", 0))
        os.environ["MASTER_PORT"] = str(_s.getsockname()[1])
os.environ.setdefault("MASTER_ADDR", "127.0.0.1")

import torch  # noqa: E402

torch.set_num_threads(1)


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an MI355X GPU")
    config.addinivalue_line("markers", "world_size(n): distributed world size")


def pytest_collection_modifyitems(config, items):
    import torch
    if not torch.cuda.is_available():
        skip_gpu = pytest.mark.skip(reason="no GPU in this container")
        for item in items:
            if "gpu" in item.keywords:
                item.add_marker(skip_gpu)
