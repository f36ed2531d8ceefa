"""Fused channels-last bias-add ops for diffusion UNets (reference:
csrc/spatial/csrc/opt_bias_add.cu via deepspeed/ops/transformer/inference/
bias_add.py:nhwc_bias_add). One entry point covers the reference's three
variants: bias-add, bias-add + residual, bias-add + residual + residual's
own bias — the HIP kernel folds all of it into one HBM pass."""

from typing import Optional

import torch

from ._loader import get_ext


def nhwc_bias_add_available(x: torch.Tensor, bias: torch.Tensor) -> bool:
    return (get_ext() is not None and x.is_cuda
            and x.dtype in (torch.bfloat16, torch.float16)
            and bias.numel() % 8 == 0 and x.numel() % bias.numel() == 0)


def nhwc_bias_add(activation: torch.Tensor, bias: torch.Tensor,
                  other: Optional[torch.Tensor] = None,
                  other_bias: Optional[torch.Tensor] = None) -> torch.Tensor:
    """activation [..., C] channels-last, bias [C]; optional residual
    `other` (same shape) and its bias `other_bias` [C]."""
    if nhwc_bias_add_available(activation, bias):
        ext = get_ext()
        return ext.nhwc_bias_add(
            activation.contiguous(), bias.contiguous(),
            other.contiguous() if other is not None else None,
            other_bias.contiguous() if other_bias is not None else None)
    out = activation + bias
    if other is not None:
        out = out + other
    if other_bias is not None:
        out = out + other_bias
    return out
