"""Checkpoint engine abstraction (reference:
deepspeed/runtime/checkpoint_engine/checkpoint_engine.py CheckpointEngine
ABC + torch_checkpoint_engine.py; the nebula engine is a config stub here).

The engine routes every checkpoint file through one of these, so
alternative storage backends (async writers, object stores) plug in
without touching the training engine.
"""

import os

import torch

from ..utils.logging import logger


class CheckpointEngine:
    def __init__(self, config_params=None):
        pass

    def create(self, tag):
        """Signal the start of a new checkpoint under `tag`."""

    def save(self, state_dict, path: str):
        raise NotImplementedError

    def load(self, path: str, map_location=None):
        raise NotImplementedError

    def commit(self, tag) -> bool:
        """Mark `tag` complete (atomic-visibility point)."""
        return True

    def makedirs(self, path, exist_ok=True):
        os.makedirs(path, exist_ok=exist_ok)


class TorchCheckpointEngine(CheckpointEngine):
    def save(self, state_dict, path: str):
        torch.save(state_dict, path)

    def load(self, path: str, map_location=None):
        return torch.load(path, map_location=map_location, weights_only=False)


class AsyncTorchCheckpointEngine(TorchCheckpointEngine):
    """Serialize on the calling thread, write bytes through the native aio
    thread pool; commit() drains. Keeps step() out of the disk path."""

    def __init__(self, config_params=None, block_size=1 << 22, n_threads=4):
        super().__init__(config_params)
        import io
        self._io = io
        from ..ops._loader import get_ext
        ext = get_ext()
        self._handle = (ext.AioHandle(block_size, n_threads)
                        if ext is not None and hasattr(ext, "AioHandle")
                        else None)

    def save(self, state_dict, path: str):
        if self._handle is None:
            return super().save(state_dict, path)
        buf = self._io.BytesIO()
        torch.save(state_dict, buf)
        data = torch.frombuffer(bytearray(buf.getvalue()), dtype=torch.uint8)
        self._handle.async_pwrite(data, path)

    def commit(self, tag) -> bool:
        if self._handle is not None:
            errors = self._handle.wait()
            if errors:
                logger.error(f"async checkpoint '{tag}': {errors} IO errors")
                return False
        return True


def create_checkpoint_engine(name: str = "torch", config_params=None):
    return {"torch": TorchCheckpointEngine,
            "async": AsyncTorchCheckpointEngine}[name](config_params)
