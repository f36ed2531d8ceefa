"""Falcon-family decoder (reference:
deepspeed/inference/v2/model_implementations/falcon — parallel
attention+MLP blocks, multi-query / grouped-query attention, RoPE).

Falcon's distinguishing block shape: attention and the MLP both read the
SAME layer-normed input and their outputs sum into the residual in one
shot (one residual add per block instead of two). New-decoder variants
(falcon-40b+) use separate norms for the attention and MLP branches;
``parallel_attn_norms`` selects that. Reuses this framework's fused
LayerNorm, RoPE tables, and GQA sdpa routing."""

from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.norms import FusedLayerNorm
from ..ops.rope import apply_rope, rope_tables
from .llama import sdpa_gqa


@dataclass
class FalconConfig:
    vocab_size: int = 65024
    hidden_size: int = 4544
    num_layers: int = 32
    num_heads: int = 71
    num_kv_heads: int = 1          # falcon-7b is MQA
    max_seq_len: int = 2048
    rope_theta: float = 10000.0
    ln_eps: float = 1e-5
    parallel_attn_norms: bool = False  # True: separate attn/mlp norms (40b+)
    initializer_range: float = 0.02

    @property
    def head_dim(self):
        return self.hidden_size // self.num_heads


def falcon_7b():
    return FalconConfig()


def falcon_mini():
    return FalconConfig(vocab_size=512, hidden_size=64, num_layers=2,
                        num_heads=4, num_kv_heads=1, max_seq_len=128)


def falcon_mini_gqa():
    return FalconConfig(vocab_size=512, hidden_size=64, num_layers=2,
                        num_heads=4, num_kv_heads=2, max_seq_len=128,
                        parallel_attn_norms=True)


class FalconBlock(nn.Module):
    def __init__(self, cfg: FalconConfig, layer_idx: int = 0):
        super().__init__()
        h = cfg.hidden_size
        d = cfg.head_dim
        self.cfg = cfg
        self.layer_idx = layer_idx
        # explicit geometry (TP sharding patches num_heads per rank;
        # cfg.head_dim is derived from the GLOBAL head count)
        self.num_heads = cfg.num_heads
        self.num_kv_heads = cfg.num_kv_heads
        self.head_dim = cfg.head_dim
        self.ln_attn = FusedLayerNorm(h, eps=cfg.ln_eps)
        self.ln_mlp = FusedLayerNorm(h, eps=cfg.ln_eps) \
            if cfg.parallel_attn_norms else None
        self.qkv = nn.Linear(
            h, (cfg.num_heads + 2 * cfg.num_kv_heads) * d, bias=False)
        self.dense = nn.Linear(cfg.num_heads * d, h, bias=False)
        self.mlp_fc = nn.Linear(h, 4 * h, bias=False)
        self.mlp_proj = nn.Linear(4 * h, h, bias=False)

    def forward(self, x, cos, sin, positions=None, kv_cache=None):
        B, S, _ = x.shape
        d = self.head_dim
        a_in = self.ln_attn(x)
        m_in = self.ln_mlp(x) if self.ln_mlp is not None else a_in

        qkv = self.qkv(a_in)
        nq = self.num_heads * d
        nk = self.num_kv_heads * d
        q = qkv[..., :nq].view(B, S, self.num_heads, d)
        k = qkv[..., nq:nq + nk].view(B, S, self.num_kv_heads, d)
        v = qkv[..., nq + nk:].view(B, S, self.num_kv_heads, d)
        q = apply_rope(q, cos, sin, positions)
        k = apply_rope(k, cos, sin, positions)
        q, k, v = (t.transpose(1, 2) for t in (q, k, v))
        attn_mask = None
        if kv_cache is not None:
            k, v = kv_cache.update(self.layer_idx, k, v)
            attn_mask = getattr(kv_cache, "last_mask", None)
        o = sdpa_gqa(q, k, v,
                     causal=attn_mask is None and q.size(2) == k.size(2),
                     attn_mask=attn_mask)
        attn_out = self.dense(o.transpose(1, 2).reshape(B, S, -1))

        mlp_out = self.mlp_proj(F.gelu(self.mlp_fc(m_in)))
        # parallel block: ONE residual add for both branches
        return x + attn_out + mlp_out


class FalconModel(nn.Module):
    def __init__(self, cfg: FalconConfig):
        super().__init__()
        self.cfg = cfg
        self.wte = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.blocks = nn.ModuleList(
            [FalconBlock(cfg, i) for i in range(cfg.num_layers)])
        self.ln_f = FusedLayerNorm(cfg.hidden_size, eps=cfg.ln_eps)
        cos, sin = rope_tables(cfg.head_dim, cfg.max_seq_len,
                               theta=cfg.rope_theta)
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)

    def forward(self, input_ids, positions=None, kv_cache=None):
        x = self.wte(input_ids)
        cos = self.rope_cos.to(x.device)
        sin = self.rope_sin.to(x.device)
        for b in self.blocks:
            x = b(x, cos, sin, positions, kv_cache=kv_cache)
        return self.ln_f(x)


class FalconForCausalLM(nn.Module):
    def __init__(self, cfg: FalconConfig):
        super().__init__()
        self.cfg = cfg
        self.transformer = FalconModel(cfg)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        self.apply(self._init_weights)

    def _init_weights(self, module):
        if isinstance(module, (nn.Linear, nn.Embedding)):
            module.weight.data.normal_(0.0, self.cfg.initializer_range)
            if getattr(module, "bias", None) is not None:
                module.bias.data.zero_()

    def forward(self, input_ids, labels=None, positions=None,
                kv_cache=None):
        hidden = self.transformer(input_ids, positions, kv_cache=kv_cache)
        if labels is None:
            return self.lm_head(hidden)
        from .llama import chunked_cross_entropy
        return chunked_cross_entropy(hidden, self.lm_head, labels)
