"""OptimizedLinear — LoRA + quantized-base linear (reference:
deepspeed/linear/optimized_linear.py, config.py, quantization.py).

Memory-efficient finetuning on MI355X: the frozen base weight is stored
groupwise-int8 (the framework quantizer, half the bytes of bf16; dequant
fuses into one kernel launch per forward) and the trainable delta is a
rank-``r`` LoRA pair. Only the LoRA params receive gradients/optimizer
state, so ZeRO shards stay tiny.
"""

from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn

from ..ops.quantizer import dequantize, quantize


@dataclass
class LoRAConfig:
    lora_r: int = 64
    lora_alpha: float = 16.0
    base_weight_sharding: int = 1  # reserved: shard frozen base over DP


@dataclass
class QuantizationConfig:
    q_bits: int = 8
    group_size: int = 2048


class QuantizedParameter(nn.Module):
    """Frozen int8/int4 weight with on-the-fly dequant."""

    def __init__(self, weight: torch.Tensor, cfg: QuantizationConfig):
        super().__init__()
        self.cfg = cfg
        self.shape = weight.shape
        self.dtype = weight.dtype
        q, s = quantize(weight.detach(), cfg.group_size, cfg.q_bits)
        self.register_buffer("q", q)
        self.register_buffer("scales", s)

    def dequantized(self) -> torch.Tensor:
        return dequantize(self.q, self.scales, int(torch.tensor(self.shape)
                          .prod()), self.cfg.group_size, self.cfg.q_bits,
                          self.dtype).view(self.shape)


class OptimizedLinear(nn.Module):
    def __init__(self, input_dim: int, output_dim: int, bias: bool = False,
                 lora_config: Optional[LoRAConfig] = None,
                 quantization_config: Optional[QuantizationConfig] = None,
                 base_weight: Optional[torch.Tensor] = None,
                 dtype=torch.bfloat16):
        super().__init__()
        self.lora_config = lora_config or LoRAConfig()
        if base_weight is None:
            base_weight = torch.empty(output_dim, input_dim, dtype=dtype)
            nn.init.kaiming_uniform_(base_weight, a=5 ** 0.5)
        if quantization_config is not None:
            self.base = QuantizedParameter(base_weight, quantization_config)
            self.base_is_quantized = True
        else:
            w = nn.Parameter(base_weight, requires_grad=False)
            self.register_parameter("base_weight", w)
            self.base_is_quantized = False
        self.bias = nn.Parameter(torch.zeros(output_dim, dtype=dtype)) \
            if bias else None

        r = self.lora_config.lora_r
        self.scaling = self.lora_config.lora_alpha / r
        self.lora_a = nn.Parameter(torch.zeros(r, input_dim, dtype=dtype))
        self.lora_b = nn.Parameter(torch.zeros(output_dim, r, dtype=dtype))
        nn.init.kaiming_uniform_(self.lora_a, a=5 ** 0.5)
        # lora_b zero-init: the layer starts exactly at the base weight

    def full_weight(self) -> torch.Tensor:
        base = self.base.dequantized() if self.base_is_quantized \
            else self.base_weight
        return base + self.scaling * (self.lora_b @ self.lora_a)

    def forward(self, x):
        base = self.base.dequantized() if self.base_is_quantized \
            else self.base_weight
        y = nn.functional.linear(x, base, self.bias)
        # two skinny GEMMs beat materializing base+BA for large shapes
        y = y + self.scaling * nn.functional.linear(
            nn.functional.linear(x, self.lora_a), self.lora_b)
        return y
