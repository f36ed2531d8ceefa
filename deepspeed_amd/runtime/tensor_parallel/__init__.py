"""Tensor-parallel training (reference: deepspeed.tp_model_init
deepspeed/__init__.py:369 + runtime/tensor_parallel/tp_manager.py)."""

import torch

from ...parallel import groups
from .layers import (ColumnParallelLinear, RowParallelLinear,
                     shard_linear_for_training)

__all__ = ["tp_model_init", "ColumnParallelLinear", "RowParallelLinear"]


def tp_model_init(model: torch.nn.Module, tp_size: int, dtype=None):
    """Shard a built model for tensor-parallel TRAINING over the xGMI mesh.
    Creates TP groups (DP = strided complement, consumed by ZeRO), replaces
    Linears with trainable shards, returns the model."""
    from ... import comm as dist
    if not dist.is_initialized():
        dist.init_distributed()
    groups.initialize_tensor_parallel(tp_size)
    g = groups.get_tensor_parallel_group()
    r = groups.get_tensor_parallel_rank()
    n = shard_linear_for_training(model, g, r, tp_size)
    if dtype is not None:
        model.to(dtype)
    from ...utils.logging import log_dist
    log_dist(f"tp_model_init: sharded {n} linears over tp={tp_size}")
    return model
