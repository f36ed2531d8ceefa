"""HuggingFace transformers models under the engine (reference contract:
DeepSpeed's primary user path is HF Trainer/accelerate models passed to
deepspeed.initialize — tests/unit/ runtime tests use HF models heavily).

Constructed from configs (random init, no network). GPT-2 exercises tied
input/output embeddings under ZeRO-3's module-unit partitioning (lm_head
shares wte's weight, so the unit is owned once and fetched through the
shared-param path) and HF's Conv1D (transposed linear) layers."""

import copy

import pytest
import torch

from .common import run_distributed

transformers = pytest.importorskip("transformers")


def _gpt2():
    from transformers import GPT2Config, GPT2LMHeadModel
    torch.manual_seed(11)
    cfg = GPT2Config(n_layer=2, n_embd=64, n_head=2, vocab_size=128,
                     n_positions=64, bos_token_id=0, eos_token_id=0,
                     resid_pdrop=0.0, embd_pdrop=0.0, attn_pdrop=0.0,
                     attn_implementation="eager")
    return GPT2LMHeadModel(cfg)


def _llama():
    from transformers import LlamaConfig, LlamaForCausalLM
    torch.manual_seed(12)
    cfg = LlamaConfig(num_hidden_layers=2, hidden_size=64,
                      intermediate_size=128, num_attention_heads=4,
                      num_key_value_heads=2, vocab_size=128,
                      max_position_embeddings=64,
                      attn_implementation="eager")
    return LlamaForCausalLM(cfg)


def _torch_reference(model, batches, lr):
    ref = copy.deepcopy(model)
    opt = torch.optim.AdamW(ref.parameters(), lr=lr)
    losses = []
    for ids in batches:
        opt.zero_grad()
        loss = ref(ids, labels=ids).loss
        loss.backward()
        opt.step()
        losses.append(float(loss.detach()))
    return losses, ref


def _hf_worker(rank, world, arch, stage):
    import deepspeed_amd
    lr, steps = 1e-3, 4
    model = {"gpt2": _gpt2, "llama": _llama}[arch]()
    ref_model = copy.deepcopy(model)
    torch.manual_seed(5)
    batches = [torch.randint(0, 128, (2, 16)) for _ in range(steps)]

    config = {
        "train_micro_batch_size_per_gpu": 2,
        "zero_optimization": {"stage": stage, "overlap_comm": False,
                              "reduce_bucket_size": 5000},
        "optimizer": {"type": "AdamW", "params": {"lr": lr}},
    }
    engine, _, _, _ = deepspeed_amd.initialize(model=model, config=config)
    losses = []
    for ids in batches:
        out = engine(ids.to(engine.device), labels=ids.to(engine.device))
        engine.backward(out.loss)
        engine.step()
        losses.append(float(out.loss.detach()))

    ref_losses, ref = _torch_reference(ref_model, batches, lr)
    for a, b in zip(losses, ref_losses):
        assert abs(a - b) < 5e-3, (arch, stage, losses, ref_losses)
    # weight parity after training (fp32 engine => tight)
    if stage == 3:
        sd = engine.optimizer.get_full_state_dict()
        if rank == 0:
            ref_sd = ref.state_dict()
            for name, t in sd.items():
                torch.testing.assert_close(
                    t.float().cpu(), ref_sd[name].float(), atol=1e-4,
                    rtol=1e-3, msg=lambda m, n=name: f"{n}: {m}")
    else:
        for (n, p), (_, q) in zip(engine.module.named_parameters(),
                                  ref.named_parameters()):
            torch.testing.assert_close(p.float().cpu(), q.float(),
                                       atol=1e-4, rtol=1e-3,
                                       msg=lambda m, n=n: f"{n}: {m}")


@pytest.mark.parametrize("arch,stage", [("gpt2", 2), ("gpt2", 3),
                                        ("llama", 3)])
def test_hf_model_zero_parity_ws2(arch, stage):
    run_distributed(_hf_worker, world_size=2, args=(arch, stage))


def test_hf_gpt2_generate_under_engine():
    """init_inference wraps an HF model; its own generate() still runs
    (the engine forwards attribute access) and outputs match the bare
    model's greedy decode."""
    import deepspeed_amd
    model = _gpt2().eval()
    ids = torch.randint(0, 128, (1, 8))
    with torch.no_grad():
        want = model.generate(ids, max_new_tokens=8, do_sample=False,
                              pad_token_id=0)
    engine = deepspeed_amd.init_inference(model, dtype=torch.float32)
    with torch.no_grad():
        got = engine.module.generate(ids, max_new_tokens=8, do_sample=False,
                                     pad_token_id=0)
    assert torch.equal(want, got)


def test_zero_init_hf_construction_then_stage3_step():
    """`with zero.Init(): model = AutoModel...` (reference large-model entry
    path, partition_parameters.py Init:824): params come out in the target
    dtype and the model trains under a stage-3 engine."""
    from .common import run_local
    run_local(_zero_init_worker)


def _zero_init_worker(rank, world):
    import deepspeed_amd
    from deepspeed_amd.runtime.zero.partition import Init
    from transformers import GPT2Config, GPT2LMHeadModel
    with Init(dtype=torch.bfloat16):
        cfg = GPT2Config(n_layer=2, n_embd=64, n_head=2, vocab_size=128,
                         n_positions=64, bos_token_id=0, eos_token_id=0,
                         resid_pdrop=0.0, embd_pdrop=0.0, attn_pdrop=0.0,
                         attn_implementation="eager")
        model = GPT2LMHeadModel(cfg)
    assert all(p.dtype == torch.bfloat16 for p in model.parameters())
    engine, _, _, _ = deepspeed_amd.initialize(model=model, config={
        "train_micro_batch_size_per_gpu": 2,
        "zero_optimization": {"stage": 3},
        "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}},
        "bf16": {"enabled": True},
    })
    ids = torch.randint(0, 128, (2, 16), device=engine.device)
    first = None
    for _ in range(3):
        loss = engine(ids, labels=ids).loss
        engine.backward(loss)
        engine.step()
        first = first if first is not None else float(loss)
    assert float(loss) < first  # memorizing one batch must reduce loss


def _autotp_gpt2_worker(rank, world):
    """AutoTP over HF GPT-2: Conv1D layers (transposed weights) + fused
    c_attn qkv head-slicing; sharded forward must match the unsharded
    model's logits."""
    import torch.distributed as tdist
    from deepspeed_amd.inference.auto_tp import (shard_model,
                                                 shard_attention_heads)
    model = _gpt2().eval()
    ref = copy.deepcopy(model)
    ids = torch.randint(0, 128, (2, 10), generator=torch.Generator()
                        .manual_seed(3))
    with torch.no_grad():
        want = ref(ids).logits
    n = shard_model(model, None, rank, world)
    assert n == 4 * 2  # c_attn + attn.c_proj + mlp.c_fc + mlp.c_proj x2 layers
    shard_attention_heads(model, rank, world)
    with torch.no_grad():
        got = model(ids).logits
    torch.testing.assert_close(got, want, atol=2e-4, rtol=1e-4)


def test_autotp_hf_gpt2_forward_parity_ws2():
    run_distributed(_autotp_gpt2_worker, world_size=2)


@pytest.mark.parametrize("model_type", ["llama", "mistral", "qwen2"])
def test_kernel_inject_hf_arch_zoo(model_type):
    """Kernel injection across the HF decoder zoo (reference
    module_inject/containers/*): RMSNorm + SwiGLU swaps must fire on every
    arch and preserve forward numerics."""
    from transformers import AutoConfig, AutoModelForCausalLM
    from deepspeed_amd.module_inject import replace_transformer_layer, \
        HFInjectionPolicy
    from deepspeed_amd.ops.norms import RMSNorm

    cfg = AutoConfig.for_model(
        model_type, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        vocab_size=256, eos_token_id=0, pad_token_id=0)
    torch.manual_seed(0)
    model = AutoModelForCausalLM.from_config(cfg)
    model.eval()
    ids = torch.randint(0, 256, (1, 12))
    with torch.no_grad():
        want = model(ids).logits
    policy = HFInjectionPolicy()
    replace_transformer_layer(model, policy)
    n = sum(policy.injected.values())
    assert n >= 2 * cfg.num_hidden_layers, policy.injected
    assert any(isinstance(m, RMSNorm) for m in model.modules())
    with torch.no_grad():
        got = model(ids).logits
    torch.testing.assert_close(got, want, rtol=2e-3, atol=2e-3)


def test_llama3_rope_scaling_matches_hf():
    """llama3-style rope_scaling frequencies bit-match HF transformers'
    _compute_llama3_parameters; a scaled model steps fine."""
    from transformers.modeling_rope_utils import ROPE_INIT_FUNCTIONS
    from transformers import AutoConfig
    from deepspeed_amd.ops.rope import llama3_scale_freqs
    from deepspeed_amd.models import LlamaForCausalLM, llama_tiny

    cfg = AutoConfig.for_model(
        "llama", hidden_size=256, num_attention_heads=4,
        num_hidden_layers=1, vocab_size=128, rope_theta=500000.0,
        rope_scaling={"rope_type": "llama3", "factor": 8.0,
                      "low_freq_factor": 1.0, "high_freq_factor": 4.0,
                      "original_max_position_embeddings": 8192},
        max_position_embeddings=32768)
    hf_inv, _ = ROPE_INIT_FUNCTIONS["llama3"](cfg, device="cpu")
    base = 1.0 / (500000.0 ** (torch.arange(0, 64, 2,
                                            dtype=torch.float64) / 64))
    torch.testing.assert_close(llama3_scale_freqs(base).float(),
                               hf_inv.float(), rtol=1e-6, atol=1e-9)

    c = llama_tiny()
    c.rope_scaling = {"factor": 8.0, "low_freq_factor": 1.0,
                      "high_freq_factor": 4.0,
                      "original_max_position_embeddings": 64}
    m = LlamaForCausalLM(c)
    ids = torch.randint(0, 500, (1, 16))
    loss = m(ids, labels=ids)
    loss.backward()
    assert torch.isfinite(loss)
