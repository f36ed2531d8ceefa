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
