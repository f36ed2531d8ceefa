"""Llama-family decoder (Llama-3 shapes) built on the CDNA4 fused ops.

This is the framework's flagship training model for the headline benchmark
(Llama-3-8B ZeRO-3 bf16 on MI355X — BASELINE.json). Random-init weights,
synthetic data; architecture matches Llama-3 (GQA attention, RoPE
theta=500000, SwiGLU MLP, RMSNorm, tied-off lm_head).

Hot ops: RMSNorm / RoPE / SwiGLU are the hand-written HIP kernels in
``deepspeed_amd.ops``; GEMMs go through torch (hipBLASLt); attention uses
torch SDPA (flash backend on ROCm) with an explicit-math fallback.
"""

from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.norms import RMSNorm
from ..ops.rope import apply_rope, rope_tables
from ..ops.swiglu import swiglu


@dataclass
class LlamaConfig:
    vocab_size: int = 128256
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_layers: int = 32
    num_heads: int = 32
    num_kv_heads: int = 8
    max_seq_len: int = 8192
    rope_theta: float = 500000.0
    rms_eps: float = 1e-5
    tie_embeddings: bool = False
    initializer_range: float = 0.02
    attention_bias: bool = False  # qwen2-style qkv bias
    # llama3-style rope_scaling dict: {factor, low_freq_factor,
    # high_freq_factor, original_max_position_embeddings}
    rope_scaling: dict = None

    @property
    def head_dim(self):
        return self.hidden_size // self.num_heads


def llama3_8b():
    return LlamaConfig()


def llama3_1_8b():
    """Llama-3.1-8B: llama3 rope scaling unlocks 128k context on the
    3-8B geometry (frequencies verified bit-exact vs HF transformers'
    _compute_llama3_parameters)."""
    return LlamaConfig(max_seq_len=131072, rope_scaling={
        "factor": 8.0, "low_freq_factor": 1.0, "high_freq_factor": 4.0,
        "original_max_position_embeddings": 8192})


def llama3_70b():
    return LlamaConfig(hidden_size=8192, intermediate_size=28672,
                       num_layers=80, num_heads=64, num_kv_heads=8)


def llama_tiny():
    """Small config for CPU tests."""
    return LlamaConfig(vocab_size=512, hidden_size=64, intermediate_size=128,
                       num_layers=2, num_heads=4, num_kv_heads=2,
                       max_seq_len=128)


def phi3_mini():
    """Phi-3-mini 3.8B shapes (BASELINE config 5: small-model ZeRO-2 path).
    Phi-3's decoder is Llama-architecture (RoPE, SwiGLU, RMSNorm) with MHA
    (no GQA) and a 32k vocab."""
    return LlamaConfig(vocab_size=32064, hidden_size=3072,
                       intermediate_size=8192, num_layers=32, num_heads=32,
                       num_kv_heads=32, max_seq_len=4096, rope_theta=10000.0,
                       tie_embeddings=False)


def mistral_7b():
    """Mistral-7B shapes (llama architecture; the 4k-window sliding
    attention is omitted — full causal attention is a superset and the
    paged KV cache caps context by max_seq_len)."""
    return LlamaConfig(vocab_size=32000, hidden_size=4096,
                       intermediate_size=14336, num_layers=32, num_heads=32,
                       num_kv_heads=8, max_seq_len=8192, rope_theta=10000.0)


def qwen2_7b():
    """Qwen2-7B shapes: Llama architecture + qkv bias, 152k vocab
    (reference inference/v2/model_implementations/qwen_v2)."""
    return LlamaConfig(vocab_size=152064, hidden_size=3584,
                       intermediate_size=18944, num_layers=28, num_heads=28,
                       num_kv_heads=4, max_seq_len=32768,
                       rope_theta=1000000.0, attention_bias=True)


def qwen2_mini():
    return LlamaConfig(vocab_size=512, hidden_size=64, intermediate_size=128,
                       num_layers=2, num_heads=4, num_kv_heads=2,
                       max_seq_len=128, attention_bias=True)


def llama_mini():
    """~0.5B for single-GPU smoke runs."""
    return LlamaConfig(vocab_size=32000, hidden_size=1024,
                       intermediate_size=2816, num_layers=8, num_heads=16,
                       num_kv_heads=4, max_seq_len=4096)


class LlamaAttention(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        self.num_heads = cfg.num_heads
        self.num_kv_heads = cfg.num_kv_heads
        self.head_dim = cfg.head_dim
        h = cfg.hidden_size
        ab = cfg.attention_bias
        self.q_proj = nn.Linear(h, cfg.num_heads * self.head_dim, bias=ab)
        self.k_proj = nn.Linear(h, cfg.num_kv_heads * self.head_dim, bias=ab)
        self.v_proj = nn.Linear(h, cfg.num_kv_heads * self.head_dim, bias=ab)
        self.o_proj = nn.Linear(cfg.num_heads * self.head_dim, h, bias=False)

    def forward(self, x, cos, sin, positions=None, kv_cache=None,
                attention_fn=None):
        B, S, _ = x.shape
        q = self.q_proj(x).view(B, S, self.num_heads, self.head_dim)
        k = self.k_proj(x).view(B, S, self.num_kv_heads, self.head_dim)
        v = self.v_proj(x).view(B, S, self.num_kv_heads, self.head_dim)
        q = apply_rope(q, cos, sin, positions)
        k = apply_rope(k, cos, sin, positions)

        if getattr(self, "sp_group", None) is not None and kv_cache is None:
            # Ulysses: all-to-all scatters heads / gathers the sequence so
            # each rank runs full-sequence causal attention on H/P heads
            o = self._ulysses_attn(q, k, v)
            return self.o_proj(o.reshape(B, S, -1))

        if kv_cache is None and attention_fn is None:
            from ..ops.attention import (flash_attn_available, flash_attn_fwd,
                                         flash_attn_func,
                                         flash_train_available)
            if flash_attn_available(q, k):
                # hand-written CDNA4 MFMA flash kernel, BSHD layout — skips
                # the transpose entirely (inference/no-grad prefill path)
                o = flash_attn_fwd(q, k, v, causal=True)
                return self.o_proj(o.reshape(B, S, -1))
            if torch.is_grad_enabled() and flash_train_available(q, k):
                # differentiable hand-written path (fwd + dk/dv/dq kernels)
                o = flash_attn_func(q, k, v, causal=True)
                return self.o_proj(o.reshape(B, S, -1))

        if (S == 1 and kv_cache is not None
                and hasattr(kv_cache, "decode_attention")
                and attention_fn is None):
            # paged flash-decode kernel over the block table (no padded
            # KV materialization); None -> ineligible, normal path below
            o = kv_cache.decode_attention(self.layer_idx, q, k, v)
            if o is not None:
                kv_cache.last_decoded = True
                return self.o_proj(o.reshape(B, S, -1))

        # [B, H, S, D] for SDPA
        q = q.transpose(1, 2)
        k = k.transpose(1, 2)
        v = v.transpose(1, 2)
        attn_mask = None
        if kv_cache is not None:
            k, v = kv_cache.update(self.layer_idx, k, v)
            # ragged caches expose a per-sequence validity mask (slots in
            # one batch sit at different positions)
            attn_mask = getattr(kv_cache, "last_mask", None)
        if attention_fn is not None:
            o = attention_fn(q, k, v)
        else:
            # prefill (q spans the whole kv prefix) is causal; decode steps
            # (q_len < kv_len) attend to the full cached prefix
            o = sdpa_gqa(q, k, v,
                         causal=attn_mask is None and q.size(2) == k.size(2),
                         attn_mask=attn_mask)
        o = o.transpose(1, 2).reshape(B, S, -1)
        return self.o_proj(o)


def _bshd_causal_attention(q, k, v):
    """Local attention in [b, s, H, d] layout (Ulysses local_attn)."""
    q, k, v = (t.transpose(1, 2) for t in (q, k, v))
    return sdpa_gqa(q, k, v, causal=True).transpose(1, 2)


def enable_ulysses(model, sp_group=None):
    """Wire Ulysses sequence parallelism into every attention module.
    Requires num_heads % sp_world == 0 and num_kv_heads % sp_world == 0."""
    from ..parallel import groups
    from ..sequence import DistributedAttention
    sp_group = sp_group or groups.get_sequence_parallel_group()
    assert sp_group is not None, "initialize_sequence_parallel first"
    import torch.distributed as td
    P = td.get_world_size(sp_group)
    for mod in model.modules():
        if isinstance(mod, LlamaAttention):
            assert mod.num_heads % P == 0 and mod.num_kv_heads % P == 0, \
                f"heads {mod.num_heads}/{mod.num_kv_heads} not divisible " \
                f"by sp={P}"
            mod.sp_group = sp_group
            mod._ulysses_attn = DistributedAttention(_bshd_causal_attention,
                                                     sp_group)
    return model


def sdpa_gqa(q, k, v, causal=True, attn_mask=None):
    """SDPA with grouped-query support (expands KV if enable_gqa missing)."""
    try:
        return F.scaled_dot_product_attention(q, k, v, attn_mask=attn_mask,
                                              is_causal=causal,
                                              enable_gqa=True)
    except (TypeError, RuntimeError):
        rep = q.shape[1] // k.shape[1]
        if rep > 1:
            k = k.repeat_interleave(rep, dim=1)
            v = v.repeat_interleave(rep, dim=1)
        return F.scaled_dot_product_attention(q, k, v, attn_mask=attn_mask,
                                              is_causal=causal)


class LlamaMLP(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        h, i = cfg.hidden_size, cfg.intermediate_size
        self.gate_proj = nn.Linear(h, i, bias=False)
        self.up_proj = nn.Linear(h, i, bias=False)
        self.down_proj = nn.Linear(i, h, bias=False)

    def forward(self, x):
        return self.down_proj(swiglu(self.gate_proj(x), self.up_proj(x)))


class LlamaDecoderLayer(nn.Module):
    def __init__(self, cfg: LlamaConfig, layer_idx: int):
        super().__init__()
        self.input_layernorm = RMSNorm(cfg.hidden_size, eps=cfg.rms_eps)
        self.self_attn = LlamaAttention(cfg)
        self.self_attn.layer_idx = layer_idx
        self.post_attention_layernorm = RMSNorm(cfg.hidden_size, eps=cfg.rms_eps)
        self.mlp = LlamaMLP(cfg)

    def forward(self, x, cos, sin, positions=None, kv_cache=None):
        x = x + self.self_attn(self.input_layernorm(x), cos, sin, positions,
                               kv_cache)
        x = x + self.mlp(self.post_attention_layernorm(x))
        return x


class LlamaModel(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(
            [LlamaDecoderLayer(cfg, i) for i in range(cfg.num_layers)])
        self.norm = RMSNorm(cfg.hidden_size, eps=cfg.rms_eps)
        self.gradient_checkpointing = False

    def forward(self, input_ids, positions=None, kv_cache=None):
        x = self.embed_tokens(input_ids)
        cos, sin = rope_tables(self.cfg.head_dim, self.cfg.max_seq_len,
                               self.cfg.rope_theta, device=x.device,
                               rope_scaling=self.cfg.rope_scaling)
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

    def gradient_checkpointing_enable(self):
        self.gradient_checkpointing = True


class LlamaForCausalLM(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        self.model = LlamaModel(cfg)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        if cfg.tie_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight
        self.apply(self._init_weights)

    def _init_weights(self, module):
        std = self.cfg.initializer_range
        if isinstance(module, nn.Linear):
            module.weight.data.normal_(0.0, std)
            if module.bias is not None:
                module.bias.data.zero_()
        elif isinstance(module, nn.Embedding):
            module.weight.data.normal_(0.0, std)

    def forward(self, input_ids, labels=None, positions=None, kv_cache=None):
        hidden = self.model(input_ids, positions, kv_cache)
        if labels is not None:
            # chunked fp32 cross-entropy: avoids materializing the full
            # [B*S, vocab] fp32 logits (vocab=128256 -> ~2 GB per 4k tokens).
            # Routed through the lm_head MODULE so ZeRO-3 fetch hooks fire.
            return chunked_cross_entropy(hidden, self.lm_head, labels)
        return self.lm_head(hidden)

    def num_parameters(self):
        return sum(p.numel() for p in self.parameters())


def chunked_cross_entropy(hidden, lm_head, labels, chunk_tokens=8192):
    """loss = CE(lm_head(hidden), labels), computed in token chunks so the
    fp32 logits never exceed chunk_tokens x vocab. With DS_AMD_FUSED_CE=1
    the per-chunk softmax+nll runs as the fused bf16 HIP kernel and no
    fp32 logits are materialized at all."""
    from ..ops.cross_entropy import fused_cross_entropy_sum
    B, S, H = hidden.shape
    hidden = hidden.reshape(-1, H)
    labels = labels.reshape(-1)
    n = hidden.shape[0]
    total = hidden.new_zeros((), dtype=torch.float32)
    count = hidden.new_zeros((), dtype=torch.float32)
    for s in range(0, n, chunk_tokens):
        e = min(s + chunk_tokens, n)
        logits = lm_head(hidden[s:e])
        l, c = fused_cross_entropy_sum(logits, labels[s:e])
        total = total + l
        count = count + c
    return total / count.clamp(min=1)
