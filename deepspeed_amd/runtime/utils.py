"""Runtime helpers: flattening, norms, clipping, memory reporting.

Capability parity with the reference's ``deepspeed/runtime/utils.py``
(global-norm/clip helpers :305/:315, see_memory_usage :771,
all_gather_dp_groups :965). Flattening uses torch's C++
_flatten_dense_tensors; partition padding aligns shards to 128 elements so
RCCL reduce-scatter segments stay 256-byte aligned (xGMI packetization).
"""

import gc
import math
from typing import List

import torch
from torch._utils import _flatten_dense_tensors, _unflatten_dense_tensors

from .. import accel
from .. import comm as dist
from ..utils.logging import logger

ALIGNMENT = 128  # elements; 256 B for bf16 — keeps RCCL segments aligned


def flatten_tensors(tensors: List[torch.Tensor]) -> torch.Tensor:
    return _flatten_dense_tensors(tensors)


def unflatten_tensors(flat: torch.Tensor, tensors: List[torch.Tensor]):
    return _unflatten_dense_tensors(flat, tensors)


def padded_size(numel: int, world_size: int, alignment: int = ALIGNMENT) -> int:
    """Total size so that numel splits evenly into world_size aligned shards."""
    chunk = alignment * world_size
    return math.ceil(numel / chunk) * chunk


def get_global_norm_of_tensors(tensors, norm_type=2.0, group=None,
                               use_foreach=True) -> torch.Tensor:
    """L2 (or inf) norm over local tensors, reduced across ``group``.

    Returns a 0-dim fp32 tensor on the tensors' device.
    """
    if len(tensors) == 0:
        total = torch.zeros((), dtype=torch.float32, device=accel.current_device())
    elif norm_type == math.inf:
        total = torch.max(torch.stack([t.detach().abs().max().float() for t in tensors]))
        if dist.is_initialized() and dist.get_world_size(group) > 1:
            dist.all_reduce(total, op=dist.ReduceOp.MAX, group=group)
        return total
    else:
        if use_foreach:
            norms = torch._foreach_norm([t.detach() for t in tensors], 2.0)
            total = torch.stack([n.float() for n in norms]).pow(2).sum()
        else:
            total = sum(t.detach().float().pow(2).sum() for t in tensors)
    if dist.is_initialized() and dist.get_world_size(group) > 1:
        dist.all_reduce(total, op=dist.ReduceOp.SUM, group=group)
    return total.sqrt()


def clip_tensors_by_global_norm(tensors, max_norm: float, global_norm: torch.Tensor,
                                eps: float = 1e-6):
    """Scale tensors in-place by max_norm / max(global_norm, max_norm)."""
    clip_coef = max_norm / (global_norm + eps)
    clip_coef = torch.clamp(clip_coef, max=1.0)
    torch._foreach_mul_(tensors, clip_coef)
    return tensors


def is_model_parallel_parameter(p) -> bool:
    return getattr(p, "model_parallel", False) or getattr(p, "tensor_model_parallel", False)


def see_memory_usage(message: str, force: bool = False, ranks=(0,)):
    if not force:
        return
    if dist.get_rank() not in ranks:
        return
    gc.collect()
    if accel.available():
        stats = accel.memory_stats()
        logger.info(
            f"{message} | MA {stats['allocated']/2**30:.2f} GB "
            f"Max_MA {stats['max_allocated']/2**30:.2f} GB "
            f"CA {stats['reserved']/2**30:.2f} GB "
            f"Max_CA {stats['max_reserved']/2**30:.2f} GB")
        accel.reset_peak_memory_stats()
    else:
        try:
            import psutil
            vm = psutil.virtual_memory()
            logger.info(f"{message} | CPU mem used {vm.used/2**30:.2f} GB "
                        f"({vm.percent:.1f}%)")
        except ImportError:
            logger.info(f"{message} | (no memory stats available)")


def empty_cache():
    if accel.available():
        torch.cuda.empty_cache()


class DummyOptim(torch.optim.Optimizer):
    """Placeholder optimizer when the user trains without one (inference/
    eval, or a parameterless pipeline stage)."""

    def __init__(self, params):
        params = list(params)
        if not params:  # torch rejects an empty parameter list
            params = [torch.nn.Parameter(torch.zeros(1))]
        super().__init__(params, defaults={"lr": 0.0})

    def step(self, closure=None):
        pass


def call_to_str(base, *args, **kwargs):
    name = f"{base}("
    if args:
        name += ", ".join(repr(a) for a in args)
        if kwargs:
            name += ", "
    if kwargs:
        name += ", ".join(f"{k}={v!r}" for k, v in kwargs.items())
    name += ")"
    return name
