"""Groupwise int8/int4 quantization ops (reference:
deepspeed/ops/quantizer + csrc/quantization — see quantize.hip for the
CDNA4 kernels). The torch fallback defines the semantics; the HIP kernel
must match it exactly (symmetric, round-to-nearest-even, scale=absmax/qmax
per group) — tests/test_quantizer.py asserts bit-equality on GPU.
"""

from typing import Tuple

import torch

from ._loader import get_ext


def quantize(x: torch.Tensor, group_size: int = 2048,
             bits: int = 8) -> Tuple[torch.Tensor, torch.Tensor]:
    """x (any float dtype, flat or not) -> (int8 buffer, fp32 scales).
    int4 packs two values per byte (lo nibble = even index)."""
    assert bits in (4, 8)
    ext = get_ext()
    flat = x.contiguous().view(-1)
    if ext is not None and x.is_cuda:
        return ext.groupwise_quant(flat, group_size, bits)
    return _torch_quantize(flat, group_size, bits)


def dequantize(q: torch.Tensor, scales: torch.Tensor, numel: int,
               group_size: int = 2048, bits: int = 8,
               dtype: torch.dtype = torch.bfloat16) -> torch.Tensor:
    ext = get_ext()
    if ext is not None and q.is_cuda:
        return ext.groupwise_dequant(q, scales, numel, group_size, bits, dtype)
    return _torch_dequantize(q, scales, numel, group_size, bits, dtype)


def _torch_quantize(flat, group_size, bits):
    n = flat.numel()
    qmax = 127.0 if bits == 8 else 7.0
    groups = (n + group_size - 1) // group_size
    padded = flat.float()
    if groups * group_size != n:
        padded = torch.cat([padded, padded.new_zeros(groups * group_size - n)])
    gview = padded.view(groups, group_size)
    amax = gview.abs().amax(dim=1)
    scales = torch.where(amax > 0, amax / qmax, torch.ones_like(amax))
    qv = torch.round(gview / scales[:, None]).clamp(-qmax, qmax).to(torch.int8)
    qv = qv.view(-1)[:n]
    if bits == 4:
        if n % 2:
            qv = torch.cat([qv, qv.new_zeros(1)])
        pairs = qv.view(-1, 2).to(torch.int16)
        packed = ((pairs[:, 1] & 0xF) << 4) | (pairs[:, 0] & 0xF)
        return packed.to(torch.uint8).view(torch.int8), scales
    return qv, scales


def _torch_dequantize(q, scales, numel, group_size, bits, dtype):
    if bits == 4:
        b = q.view(torch.uint8).to(torch.int16)
        lo = (b & 0xF)
        hi = (b >> 4) & 0xF
        vals = torch.stack([lo, hi], dim=1).view(-1)[:numel]
        vals = torch.where(vals >= 8, vals - 16, vals).float()
    else:
        vals = q.float()[:numel]
    idx = torch.arange(numel, device=q.device) // group_size
    return (vals * scales[idx]).to(dtype)
