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
