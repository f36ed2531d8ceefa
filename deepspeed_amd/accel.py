"""Thin ROCm/MI355X accelerator helpers.

The reference maintains an N-backend accelerator abstraction
(``accelerator/abstract_accelerator.py`` with ~80 abstract methods). This
framework targets exactly one accelerator — AMD Instinct MI355X (gfx950)
under ROCm — so these are plain functions over ``torch.cuda`` (which is the
HIP runtime on ROCm builds), with a CPU fallback so the full control-plane
test suite runs on GPU-less hosts.
"""

import os
import functools

import torch

GFX_ARCH = "gfx950"


def available() -> bool:
    return torch.cuda.is_available()


@functools.lru_cache(None)
def device_name() -> str:
    if available():
        return torch.cuda.get_device_name(0)
    return "cpu"


def current_device() -> torch.device:
    if available():
        return torch.device("cuda", torch.cuda.current_device())
    return torch.device("cpu")


def device_for(local_rank: int) -> torch.device:
    if available():
        return torch.device("cuda", local_rank)
    return torch.device("cpu")


def set_device(local_rank: int) -> None:
    if available():
        torch.cuda.set_device(local_rank)


def synchronize() -> None:
    if available():
        torch.cuda.synchronize()


def communication_backend_name() -> str:
    """'nccl' is RCCL on ROCm builds of PyTorch."""
    return "nccl" if available() else "gloo"


def stream(priority: int = 0):
    if available():
        return torch.cuda.Stream(priority=priority)
    return _NullStream()


def current_stream():
    if available():
        return torch.cuda.current_stream()
    return _NullStream()


def stream_ctx(s):
    if available() and isinstance(s, torch.cuda.Stream):
        return torch.cuda.stream(s)
    import contextlib
    return contextlib.nullcontext()


def event(enable_timing: bool = False):
    if available():
        return torch.cuda.Event(enable_timing=enable_timing)
    return _NullEvent()


class _NullStream:
    def wait_stream(self, other):
        pass

    def wait_event(self, ev):
        pass

    def synchronize(self):
        pass

    def record_event(self, ev=None):
        return ev if ev is not None else _NullEvent()


class _NullEvent:
    def record(self, stream=None):
        pass

    def wait(self, stream=None):
        pass

    def synchronize(self):
        pass

    def query(self):
        return True

    def elapsed_time(self, other):
        return 0.0


def memory_stats() -> dict:
    if not available():
        return {"allocated": 0, "reserved": 0, "max_allocated": 0, "max_reserved": 0}
    return {
        "allocated": torch.cuda.memory_allocated(),
        "reserved": torch.cuda.memory_reserved(),
        "max_allocated": torch.cuda.max_memory_allocated(),
        "max_reserved": torch.cuda.max_memory_reserved(),
    }


def reset_peak_memory_stats() -> None:
    if available():
        torch.cuda.reset_peak_memory_stats()


def total_memory() -> int:
    if available():
        return torch.cuda.get_device_properties(0).total_memory
    return 0


def pin_memory(tensor: torch.Tensor) -> torch.Tensor:
    if available():
        return tensor.pin_memory()
    return tensor


def local_rank_from_env() -> int:
    return int(os.environ.get("LOCAL_RANK", 0))


def supports_bf16() -> bool:
    if available():
        return torch.cuda.is_bf16_supported()
    return True  # CPU bf16 emulation works for tests


def get_accelerator():
    """Reference-API compatibility (accelerator/real_accelerator.py:51):
    returns this module — deliberately a single ROCm/MI355X implementation,
    not an N-backend ABC (SURVEY §7 design stance)."""
    import sys
    return sys.modules[__name__]


# reference DeepSpeedAccelerator method-name aliases
def device_count() -> int:
    import torch
    return torch.cuda.device_count() if available() else 0


def memory_allocated(device=None) -> int:
    import torch
    return torch.cuda.memory_allocated(device) if available() else 0


def max_memory_allocated(device=None) -> int:
    import torch
    return torch.cuda.max_memory_allocated(device) if available() else 0


def empty_cache() -> None:
    import torch
    if available():
        torch.cuda.empty_cache()
