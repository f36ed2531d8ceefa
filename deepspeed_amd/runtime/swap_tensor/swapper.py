"""Tensor swapping to NVMe/disk (reference: deepspeed/runtime/swap_tensor/
partitioned_param_swapper.py AsyncPartitionedParameterSwapper :37 +
optimizer swappers over the csrc/aio engine).

Keyed async swap-out/swap-in of host tensors through the C++ thread-pool
I/O engine (ops/csrc/aio.cpp). The caller owns placement policy (what to
evict when); this class owns the file lifecycle and overlap."""

import os
from typing import Dict

import torch

from ...ops._loader import get_ext


class AsyncTensorSwapper:
    def __init__(self, swap_dir: str, block_size: int = 1 << 20,
                 n_threads: int = 8):
        ext = get_ext()
        if ext is None or not hasattr(ext, "AioHandle"):
            raise RuntimeError("AsyncTensorSwapper needs the native aio op")
        self.swap_dir = swap_dir
        os.makedirs(swap_dir, exist_ok=True)
        self.handle = ext.AioHandle(block_size, n_threads)
        self._meta: Dict[str, tuple] = {}
        self._pending = 0

    def _path(self, key: str) -> str:
        return os.path.join(self.swap_dir, f"{key}.swp")

    def swap_out(self, key: str, tensor: torch.Tensor):
        """Asynchronously persist a host tensor; caller may free it AFTER
        synchronize()."""
        t = tensor.detach().contiguous()
        self._meta[key] = (t.shape, t.dtype)
        self.handle.async_pwrite(t, self._path(key))
        self._pending += 1

    def swap_in(self, key: str, out: torch.Tensor = None) -> torch.Tensor:
        """Asynchronously read a previously swapped tensor. Valid after
        synchronize()."""
        shape, dtype = self._meta[key]
        if out is None:
            out = torch.empty(shape, dtype=dtype)
        assert out.shape == shape and out.dtype == dtype, "swap_in mismatch"
        self.handle.async_pread(out.contiguous(), self._path(key))
        self._pending += 1
        return out

    def synchronize(self):
        if self._pending:
            errors = self.handle.wait()
            self._pending = 0
            if errors:
                raise IOError(f"tensor swapper: {errors} chunk IO errors")

    def remove(self, key: str):
        self._meta.pop(key, None)
        try:
            os.unlink(self._path(key))
        except FileNotFoundError:
            pass
