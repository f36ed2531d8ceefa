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


# ---------------------------------------------------------------- FP4/6/12
# Hand-written HIP kernel path (csrc/fp_quant.hip): tight bit-packing
# (FP6: 4 values -> 3 bytes), per-group fp32 scales, RNE conversion.
# Formats match the reference mapping (quantize.py:71-77):
#   4 -> e2m1, 6 -> e3m2, 8 -> e4m3, 12 -> e7m4

_FMT = {4: (2, 1), 6: (3, 2), 8: (4, 3), 12: (7, 4)}


def fp_quantize(x: torch.Tensor, bits: int = 6, group_size: int = 2048):
    """x (float, cuda) -> (packed uint8, fp32 scales). GPU kernel."""
    from ._loader import get_ext
    ext = get_ext()
    assert ext is not None and x.is_cuda, "fp_quantize needs the HIP ext"
    out, scales = ext.fp_quantize(x.contiguous().view(-1), group_size, bits)
    return out, scales


def fp_dequantize(q: torch.Tensor, scales: torch.Tensor, numel: int,
                  bits: int = 6, group_size: int = 2048,
                  out_dtype: torch.dtype = torch.bfloat16) -> torch.Tensor:
    from ._loader import get_ext
    ext = get_ext()
    return ext.fp_dequantize(q, scales, numel, group_size, bits, out_dtype)


def fp_emulate_reference(x: torch.Tensor, bits: int = 6,
                         group_size: int = 2048) -> torch.Tensor:
    """Bit-accurate torch emulation of quantize->dequantize (any device):
    the numerics contract the HIP kernel is tested against."""
    E, M = _FMT[bits]
    bias = (1 << (E - 1)) - 1
    emax = (1 << E) - 2 - bias
    emin = 1 - bias
    qmax = (2.0 - 2.0 ** (-M)) * (2.0 ** emax)
    flat = x.float().contiguous().view(-1)
    n = flat.numel()
    groups = (n + group_size - 1) // group_size
    pad = groups * group_size - n
    if pad:
        flat = torch.cat([flat, flat.new_zeros(pad)])
    g = flat.view(groups, group_size)
    amax = g.abs().amax(dim=1, keepdim=True)
    scale = torch.where(amax > 0, amax / qmax, torch.ones_like(amax))
    y = g * (1.0 / scale)  # reciprocal-multiply, matching the kernel
    # quantize each value to the e/m grid with RNE
    mag = y.abs()
    expo = torch.floor(torch.log2(mag.clamp_min(1e-45)))
    expo = expo.clamp(min=emin)                    # subnormal quantum floor
    quantum = torch.exp2(expo - M)
    q = torch.round(mag / quantum) * quantum       # RNE (torch.round)
    q = q.clamp(max=qmax) * y.sign()
    return (q * scale).view(-1)[:n].view(x.shape)
