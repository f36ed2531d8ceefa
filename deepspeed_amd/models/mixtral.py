"""Mixtral-family decoder: Llama backbone with top-2 MoE FFN per layer.

Benchmark config 3 in BASELINE.json: "Mixtral 8x7B with expert-parallel
all-to-all over xGMI". Random-init weights, synthetic data. The MoE layer
is ``deepspeed_amd.moe.MoE`` (EP all-to-all dispatch); attention/norm/RoPE
are the same CDNA4 ops as the Llama model.
"""

from dataclasses import dataclass

import torch
import torch.nn as nn

from ..moe import MoE
from ..ops.norms import RMSNorm
from ..ops.rope import rope_tables
from .llama import LlamaAttention, LlamaMLP, LlamaConfig, chunked_cross_entropy


@dataclass
class MixtralConfig:
    vocab_size: int = 32000
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_layers: int = 32
    num_heads: int = 32
    num_kv_heads: int = 8
    max_seq_len: int = 8192
    rope_theta: float = 1000000.0
    rms_eps: float = 1e-5
    num_experts: int = 8
    top_k: int = 2
    ep_size: int = 1
    capacity_factor: float = 1.25
    min_capacity: int = 4
    aux_loss_coef: float = 0.02
    initializer_range: float = 0.02

    @property
    def head_dim(self):
        return self.hidden_size // self.num_heads

    def as_llama(self) -> LlamaConfig:
        return LlamaConfig(vocab_size=self.vocab_size,
                           hidden_size=self.hidden_size,
                           intermediate_size=self.intermediate_size,
                           num_layers=self.num_layers,
                           num_heads=self.num_heads,
                           num_kv_heads=self.num_kv_heads,
                           max_seq_len=self.max_seq_len,
                           rope_theta=self.rope_theta,
                           rms_eps=self.rms_eps)


def mixtral_8x7b(ep_size=1):
    return MixtralConfig(ep_size=ep_size)


def mixtral_tiny(ep_size=1, num_experts=4):
    return MixtralConfig(vocab_size=512, hidden_size=64, intermediate_size=128,
                         num_layers=2, num_heads=4, num_kv_heads=2,
                         max_seq_len=128, num_experts=num_experts,
                         ep_size=ep_size)


def mixtral_mini(ep_size=1):
    return MixtralConfig(vocab_size=32000, hidden_size=1024,
                         intermediate_size=2816, num_layers=8, num_heads=16,
                         num_kv_heads=4, max_seq_len=4096, num_experts=8,
                         ep_size=ep_size)


class MixtralDecoderLayer(nn.Module):
    def __init__(self, cfg: MixtralConfig, layer_idx: int):
        super().__init__()
        lc = cfg.as_llama()
        self.input_layernorm = RMSNorm(cfg.hidden_size, eps=cfg.rms_eps)
        self.self_attn = LlamaAttention(lc)
        self.self_attn.layer_idx = layer_idx
        self.post_attention_layernorm = RMSNorm(cfg.hidden_size, eps=cfg.rms_eps)
        self.block_sparse_moe = MoE(
            hidden_size=cfg.hidden_size,
            expert=LlamaMLP(lc),
            num_experts=cfg.num_experts,
            ep_size=cfg.ep_size,
            k=cfg.top_k,
            capacity_factor=cfg.capacity_factor,
            eval_capacity_factor=cfg.capacity_factor,
            min_capacity=cfg.min_capacity)

    def forward(self, x, cos, sin, positions=None, kv_cache=None):
        x = x + self.self_attn(self.input_layernorm(x), cos, sin, positions,
                               kv_cache)
        moe_out, l_aux, _ = self.block_sparse_moe(
            self.post_attention_layernorm(x))
        self.l_aux = l_aux
        return x + moe_out


class MixtralModel(nn.Module):
    def __init__(self, cfg: MixtralConfig):
        super().__init__()
        self.cfg = cfg
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(
            [MixtralDecoderLayer(cfg, i) for i in range(cfg.num_layers)])
        self.norm = RMSNorm(cfg.hidden_size, eps=cfg.rms_eps)
        self.gradient_checkpointing = False

    def gradient_checkpointing_enable(self):
        self.gradient_checkpointing = True

    def forward(self, input_ids, positions=None, kv_cache=None):
        x = self.embed_tokens(input_ids)
        cos, sin = rope_tables(self.cfg.head_dim, self.cfg.max_seq_len,
                               self.cfg.rope_theta, device=x.device)
        recompute = (self.gradient_checkpointing and self.training
                     and torch.is_grad_enabled())
        if recompute:
            from ..runtime.activation_checkpointing import checkpoint
        for layer in self.layers:
            if recompute:
                x = checkpoint(lambda x_, l=layer: l(x_, cos, sin, positions,
                                                     kv_cache), x)
            else:
                x = layer(x, cos, sin, positions, kv_cache)
        return self.norm(x)


class MixtralForCausalLM(nn.Module):
    def __init__(self, cfg: MixtralConfig):
        super().__init__()
        self.cfg = cfg
        self.model = MixtralModel(cfg)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        self.apply(self._init_weights)

    def _init_weights(self, module):
        std = self.cfg.initializer_range
        if isinstance(module, nn.Linear):
            module.weight.data.normal_(0.0, std)
            if module.bias is not None:
                module.bias.data.zero_()
        elif isinstance(module, nn.Embedding):
            module.weight.data.normal_(0.0, std)
        else:
            from ..moe.experts import FusedExperts
            if isinstance(module, FusedExperts):
                module.reset_parameters(std)

    def aux_loss(self):
        terms = [layer.l_aux for layer in self.model.layers
                 if hasattr(layer, "l_aux")]
        if not terms:
            return torch.tensor(0.0)
        return torch.stack([t.float() for t in terms]).sum()

    def forward(self, input_ids, labels=None, positions=None, kv_cache=None):
        hidden = self.model(input_ids, positions, kv_cache)
        if labels is not None:
            ce = chunked_cross_entropy(hidden, self.lm_head, labels)
            return ce + self.cfg.aux_loss_coef * self.aux_loss().to(ce.device)
        return self.lm_head(hidden)

    def num_parameters(self):
        return sum(p.numel() for p in self.parameters())
