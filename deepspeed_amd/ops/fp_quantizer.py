"""FP8 (E4M3) groupwise quantization (reference: deepspeed/ops/fp_quantizer
+ csrc/fp_quantizer/fp_quantize.cu FP6/FP8/FP12).

CDNA4 has native FP8 (E4M3/E5M2) datapaths and torch-ROCm exposes
float8_e4m3fn/e5m2 tensor dtypes whose casts lower to the HW conversion
instructions — so unlike the int4/int8 quantizer (quantize.hip) this needs
no custom kernel: the groupwise scale + cast IS the fused op. Used for
fp8 gradient/weight communication experiments and MoQ-style checkpoints.
"""

from typing import Tuple

import torch

_FP8_MAX = {torch.float8_e4m3fn: 448.0, torch.float8_e5m2: 57344.0}


def fp8_quantize(x: torch.Tensor, group_size: int = 2048,
                 dtype: torch.dtype = torch.float8_e4m3fn
                 ) -> Tuple[torch.Tensor, torch.Tensor]:
    """x (float) -> (fp8 tensor, fp32 scales); groupwise symmetric."""
    flat = x.contiguous().view(-1).float()
    n = flat.numel()
    groups = (n + group_size - 1) // group_size
    if groups * group_size != n:
        flat = torch.cat([flat, flat.new_zeros(groups * group_size - n)])
    g = flat.view(groups, group_size)
    amax = g.abs().amax(dim=1)
    qmax = _FP8_MAX[dtype]
    scales = torch.where(amax > 0, amax / qmax, torch.ones_like(amax))
    q = (g / scales[:, None]).to(dtype).view(-1)[:n]
    return q, scales


def fp8_dequantize(q: torch.Tensor, scales: torch.Tensor, numel: int,
                   group_size: int = 2048,
                   out_dtype: torch.dtype = torch.bfloat16) -> torch.Tensor:
    idx = torch.arange(numel, device=q.device) // group_size
    return (q.float()[:numel] * scales[idx]).to(out_dtype)


class FP8Quantizer:
    """Stateful wrapper matching the reference FP_Quantize interface."""

    def __init__(self, group_size: int = 2048, q_bits: int = 8):
        assert q_bits == 8, "only FP8 supported (FP6/FP12 deferred)"
        self.group_size = group_size

    def quantize(self, x, q_bits: int = 8, return_meta_tensor: bool = True):
        q, s = fp8_quantize(x, self.group_size)
        self._shape = x.shape
        self._dtype = x.dtype if x.dtype.is_floating_point else torch.bfloat16
        return (q, s) if return_meta_tensor else q

    def dequantize(self, q, scale=None, q_bits: int = 8):
        n = int(torch.tensor(self._shape).prod())
        return fp8_dequantize(q, scale, n, self.group_size,
                              self._dtype).view(self._shape)
