"""DeepSpeedTransformerLayer — the classic fused encoder-layer API
(reference: deepspeed/ops/transformer/transformer.py
DeepSpeedTransformerLayer + DeepSpeedTransformerConfig over
csrc/transformer).

Same constructor contract as the reference; the computation is the
framework's fused ops (FusedLayerNorm HIP kernel, SDPA attention,
hipBLASLt GEMMs) in either pre- or post-LayerNorm arrangement.
"""

from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F

from .norms import FusedLayerNorm


@dataclass
class DeepSpeedTransformerConfig:
    batch_size: int = 1
    hidden_size: int = 1024
    intermediate_size: int = 4096
    heads: int = 16
    attn_dropout_ratio: float = 0.1
    hidden_dropout_ratio: float = 0.1
    num_hidden_layers: int = 24
    initializer_range: float = 0.02
    layer_norm_eps: float = 1e-12
    local_rank: int = -1
    seed: int = -1
    fp16: bool = False
    pre_layer_norm: bool = True


class DeepSpeedTransformerLayer(nn.Module):
    def __init__(self, config: DeepSpeedTransformerConfig,
                 initial_weights=None, initial_biases=None):
        super().__init__()
        self.config = config
        h = config.hidden_size
        self.num_heads = config.heads
        self.head_dim = h // config.heads
        if config.seed > 0:
            torch.manual_seed(config.seed)

        self.attn_qkvw = nn.Linear(h, 3 * h)
        self.attn_ow = nn.Linear(h, h)
        self.attn_nw = FusedLayerNorm(h, eps=config.layer_norm_eps)
        self.inter_w = nn.Linear(h, config.intermediate_size)
        self.output_w = nn.Linear(config.intermediate_size, h)
        self.norm_w = FusedLayerNorm(h, eps=config.layer_norm_eps)
        self.attn_dropout = nn.Dropout(config.attn_dropout_ratio)
        self.hidden_dropout = nn.Dropout(config.hidden_dropout_ratio)

        if initial_weights is not None:
            with torch.no_grad():
                for dst, src in zip(
                        (self.attn_qkvw.weight, self.attn_ow.weight,
                         self.inter_w.weight, self.output_w.weight),
                        initial_weights):
                    dst.copy_(src)
        if initial_biases is not None:
            with torch.no_grad():
                for dst, src in zip(
                        (self.attn_qkvw.bias, self.attn_ow.bias,
                         self.inter_w.bias, self.output_w.bias),
                        initial_biases):
                    dst.copy_(src)

    def _attention(self, x, attention_mask):
        B, S, H = x.shape
        qkv = self.attn_qkvw(x).view(B, S, 3, self.num_heads, self.head_dim)
        q, k, v = (qkv[:, :, i].transpose(1, 2) for i in range(3))
        if attention_mask is not None and attention_mask.dim() == 2:
            attention_mask = attention_mask[:, None, None, :].bool()
        o = F.scaled_dot_product_attention(
            q, k, v, attn_mask=attention_mask,
            dropout_p=self.config.attn_dropout_ratio if self.training else 0.0)
        return self.attn_ow(o.transpose(1, 2).reshape(B, S, H))

    def forward(self, hidden_states, attention_mask=None):
        x = hidden_states
        if self.config.pre_layer_norm:
            a = self._attention(self.attn_nw(x), attention_mask)
            x = x + self.hidden_dropout(a)
            f = self.output_w(F.gelu(self.inter_w(self.norm_w(x)),
                                     approximate="tanh"))
            return x + self.hidden_dropout(f)
        a = self._attention(x, attention_mask)
        x = self.attn_nw(x + self.hidden_dropout(a))
        f = self.output_w(F.gelu(self.inter_w(x), approximate="tanh"))
        return self.norm_w(x + self.hidden_dropout(f))
