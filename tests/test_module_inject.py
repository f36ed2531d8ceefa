"""HF kernel-injection tests (reference contract:
tests/unit/inference v1 injection over the HF zoo, offline subset)."""

import pytest
import torch

transformers = pytest.importorskip("transformers")


def _hf_llama():
    from transformers import LlamaConfig, LlamaForCausalLM
    cfg = LlamaConfig(vocab_size=256, hidden_size=64, intermediate_size=96,
                      num_hidden_layers=2, num_attention_heads=4,
                      num_key_value_heads=2, max_position_embeddings=64)
    torch.manual_seed(3)
    return LlamaForCausalLM(cfg).eval()


def test_injection_swaps_and_preserves_outputs():
    from deepspeed_amd.module_inject import (replace_transformer_layer,
                                             HFInjectionPolicy)
    m = _hf_llama()
    ids = torch.randint(0, 256, (2, 16))
    with torch.no_grad():
        ref = m(ids).logits
    policy = HFInjectionPolicy()
    replace_transformer_layer(m, policy)
    # every norm and MLP swapped
    assert policy.injected["rmsnorm"] == 2 * 2 + 1
    assert policy.injected["swiglu_mlp"] == 2
    from deepspeed_amd.ops.norms import RMSNorm
    assert isinstance(m.model.norm, RMSNorm)
    with torch.no_grad():
        got = m(ids).logits
    torch.testing.assert_close(got, ref, rtol=1e-4, atol=1e-5)


def test_injection_via_init_inference():
    import deepspeed_amd
    m = _hf_llama()
    ids = torch.randint(0, 256, (2, 16))
    with torch.no_grad():
        ref = m(ids).logits
    eng = deepspeed_amd.init_inference(m, dtype="fp32",
                                       replace_with_kernel_inject=True)
    assert eng.injection_policy.injected["swiglu_mlp"] == 2
    with torch.no_grad():
        got = eng(ids).logits
    torch.testing.assert_close(got, ref, rtol=1e-4, atol=1e-5)


@pytest.mark.gpu
def test_injection_flash_sdpa_gpu():
    """On GPU with a flash-eligible shape the injected sdpa routes to the
    MFMA kernel (counted) and matches the eager HF output."""
    from transformers import LlamaConfig, LlamaForCausalLM
    from deepspeed_amd.module_inject import (replace_transformer_layer,
                                             HFInjectionPolicy)
    cfg = LlamaConfig(vocab_size=256, hidden_size=256,
                      intermediate_size=384, num_hidden_layers=2,
                      num_attention_heads=2, num_key_value_heads=2,
                      max_position_embeddings=128,
                      attn_implementation="sdpa")
    torch.manual_seed(3)
    m = LlamaForCausalLM(cfg).eval().cuda().bfloat16()  # head_dim = 128
    ids = torch.randint(0, 256, (2, 64), device="cuda")
    with torch.no_grad():
        ref = m(ids).logits
    policy = HFInjectionPolicy()
    replace_transformer_layer(m, policy)
    with torch.no_grad():
        got = m(ids).logits
    assert policy.injected.get("flash_sdpa_calls", 0) >= 2, policy.injected
    torch.testing.assert_close(got.float(), ref.float(), rtol=5e-2, atol=5e-1)


def test_diffusers_block_injection_parity():
    """Fused diffusers transformer block vs a synthetic HF-diffusers-style
    BasicTransformerBlock (reference diffusers_transformer_block.py)."""
    import torch.nn as nn
    from deepspeed_amd.module_inject.diffusers import (
        replace_diffusers_blocks, DiffusersTransformerBlock)

    D, FF, H = 32, 64, 4

    class MiniAttn(nn.Module):
        def __init__(self):
            super().__init__()
            self.qkv = nn.Linear(D, 3 * D, bias=False)
            self.out = nn.Linear(D, D, bias=False)

        def forward(self, x, context=None):
            src = x if context is None else context
            q = self.qkv(x)[..., :D]
            k = self.qkv(src)[..., D:2 * D]
            v = self.qkv(src)[..., 2 * D:]
            B, S, _ = q.shape
            Sk = k.shape[1]
            q = q.view(B, S, H, D // H).transpose(1, 2)
            k = k.view(B, Sk, H, D // H).transpose(1, 2)
            v = v.view(B, Sk, H, D // H).transpose(1, 2)
            o = torch.nn.functional.scaled_dot_product_attention(q, k, v)
            return self.out(o.transpose(1, 2).reshape(B, S, D))

    class GEGLU(nn.Module):
        def __init__(self):
            super().__init__()
            self.proj = nn.Linear(D, 2 * FF)

        def forward(self, x):
            up, gate = self.proj(x).chunk(2, dim=-1)
            return up * torch.nn.functional.gelu(gate)

    class Block(nn.Module):
        def __init__(self):
            super().__init__()
            self.norm1 = nn.LayerNorm(D)
            self.norm2 = nn.LayerNorm(D)
            self.norm3 = nn.LayerNorm(D)
            self.attn1 = MiniAttn()
            self.attn2 = MiniAttn()
            self.ff = nn.Module()
            self.ff.net = nn.ModuleList(
                [GEGLU(), nn.Dropout(0.0), nn.Linear(FF, D)])

        def forward(self, x, context=None):
            x = self.attn1(self.norm1(x)) + x
            x = self.attn2(self.norm2(x), context) + x
            h = self.norm3(x)
            for m in self.ff.net:
                h = m(h)
            return h + x

    torch.manual_seed(5)
    model = nn.Sequential()
    model.block = Block()
    x = torch.randn(2, 9, D)
    ctx = torch.randn(2, 5, D)
    want = model.block(x, context=ctx)
    n = replace_diffusers_blocks(model)
    assert n == 1 and isinstance(model.block, DiffusersTransformerBlock)
    got = model.block(x, context=ctx)
    # tanh-approx gelu in the fused GEGLU vs diffusers' exact gelu
    torch.testing.assert_close(got, want, rtol=2e-3, atol=3e-4)
    # encoder_hidden_states kwarg spelling (diffusers >= 0.11)
    got2 = model.block(x, encoder_hidden_states=ctx)
    torch.testing.assert_close(got2, got, rtol=0, atol=0)
