"""GPT-2-family decoder (LayerNorm + GELU + learned positions).

Used for the CPU/gloo plumbing config (BASELINE.json config 1: GPT-2 small
ZeRO-1 world_size=2) and as the second model family. Hot ops route through
the fused LayerNorm / GeGLU-capable kernels.
"""

from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.norms import FusedLayerNorm
from .llama import sdpa_gqa


@dataclass
class GPT2Config:
    vocab_size: int = 50257
    hidden_size: int = 768
    num_layers: int = 12
    num_heads: int = 12
    max_seq_len: int = 1024
    ln_eps: float = 1e-5
    initializer_range: float = 0.02
    activation: str = "gelu"   # "relu" for OPT-family
    mlp_ratio: int = 4

    @property
    def head_dim(self):
        return self.hidden_size // self.num_heads


def gpt2_small():
    return GPT2Config()


def opt_125m():
    """OPT-125M shapes: GPT-2 topology with ReLU MLPs (reference
    inference/v2/model_implementations/opt)."""
    return GPT2Config(vocab_size=50272, hidden_size=768, num_layers=12,
                      num_heads=12, max_seq_len=2048, activation="relu")


def opt_mini():
    return GPT2Config(vocab_size=512, hidden_size=64, num_layers=2,
                      num_heads=4, max_seq_len=128, activation="relu")


def gpt2_tiny():
    return GPT2Config(vocab_size=512, hidden_size=64, num_layers=2,
                      num_heads=4, max_seq_len=128)


class GPT2Block(nn.Module):
    def __init__(self, cfg: GPT2Config, layer_idx: int = 0):
        super().__init__()
        h = cfg.hidden_size
        self.layer_idx = layer_idx
        self.ln_1 = FusedLayerNorm(h, eps=cfg.ln_eps)
        self.attn = nn.Linear(h, 3 * h)
        self.attn_out = nn.Linear(h, h)
        self.ln_2 = FusedLayerNorm(h, eps=cfg.ln_eps)
        self.cfg = cfg
        self.mlp_fc = nn.Linear(h, cfg.mlp_ratio * h)
        self.mlp_proj = nn.Linear(cfg.mlp_ratio * h, h)
        self.num_heads = cfg.num_heads
        self.head_dim = cfg.head_dim

    def forward(self, x, kv_cache=None):
        B, S, H = x.shape
        qkv = self.attn(self.ln_1(x))
        q, k, v = qkv.split(H, dim=-1)
        q = q.view(B, S, self.num_heads, self.head_dim).transpose(1, 2)
        k = k.view(B, S, self.num_heads, self.head_dim).transpose(1, 2)
        v = v.view(B, S, self.num_heads, self.head_dim).transpose(1, 2)
        attn_mask = None
        if kv_cache is not None:
            k, v = kv_cache.update(self.layer_idx, k, v)
            attn_mask = getattr(kv_cache, "last_mask", None)
        o = sdpa_gqa(q, k, v,
                     causal=attn_mask is None and q.size(2) == k.size(2),
                     attn_mask=attn_mask)
        o = o.transpose(1, 2).reshape(B, S, H)
        x = x + self.attn_out(o)
        h = self.mlp_fc(self.ln_2(x))
        if self.cfg.activation == "relu":
            return x + self.mlp_proj(F.relu(h))
        x = x + self.mlp_proj(F.gelu(h,
                                     approximate="tanh"))
        return x


class GPT2ForCausalLM(nn.Module):
    def __init__(self, cfg: GPT2Config):
        super().__init__()
        self.cfg = cfg
        self.wte = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.wpe = nn.Embedding(cfg.max_seq_len, cfg.hidden_size)
        self.blocks = nn.ModuleList([GPT2Block(cfg, i)
                                     for i in range(cfg.num_layers)])
        self.ln_f = FusedLayerNorm(cfg.hidden_size, eps=cfg.ln_eps)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        self.lm_head.weight = self.wte.weight  # tied
        self.apply(self._init_weights)

    def _init_weights(self, module):
        std = self.cfg.initializer_range
        if isinstance(module, (nn.Linear, nn.Embedding)):
            module.weight.data.normal_(0.0, std)
            if isinstance(module, nn.Linear) and module.bias is not None:
                module.bias.data.zero_()

    def forward(self, input_ids, labels=None, positions=None, kv_cache=None):
        B, S = input_ids.shape
        if positions is None:
            pos_emb = self.wpe(torch.arange(S, device=input_ids.device))[None]
        else:
            pos_emb = self.wpe(positions.long())
        x = self.wte(input_ids) + pos_emb
        for blk in self.blocks:
            x = blk(x, kv_cache=kv_cache)
        x = self.ln_f(x)
        logits = self.lm_head(x)
        if labels is not None:
            return F.cross_entropy(logits.float().view(-1, logits.shape[-1]),
                                   labels.view(-1), ignore_index=-100)
        return logits

    def num_parameters(self):
        return sum(p.numel() for p in self.parameters())
