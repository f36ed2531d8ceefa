"""OptimizedLinear/LoRA + sparse-gradient allreduce tests (reference
contracts: tests/unit/linear/test_linear.py, runtime sparse grads)."""

import torch

from .common import run_distributed


def test_optimized_linear_starts_at_base():
    from deepspeed_amd.linear import (LoRAConfig, OptimizedLinear,
                                      QuantizationConfig)
    torch.manual_seed(0)
    base = torch.randn(32, 16)
    lin = OptimizedLinear(16, 32, base_weight=base.clone(),
                          lora_config=LoRAConfig(lora_r=4),
                          dtype=torch.float32)
    x = torch.randn(5, 16)
    # lora_b is zero-init: output == base linear exactly
    torch.testing.assert_close(lin(x), x @ base.t(), rtol=1e-5, atol=1e-6)
    # only LoRA params are trainable
    trainable = [n for n, p in lin.named_parameters() if p.requires_grad]
    assert sorted(trainable) == ["lora_a", "lora_b"]


def test_optimized_linear_quantized_base():
    from deepspeed_amd.linear import (LoRAConfig, OptimizedLinear,
                                      QuantizationConfig)
    torch.manual_seed(1)
    base = torch.randn(64, 32)
    lin = OptimizedLinear(32, 64, base_weight=base.clone(),
                          lora_config=LoRAConfig(lora_r=8),
                          quantization_config=QuantizationConfig(
                              q_bits=8, group_size=256),
                          dtype=torch.float32)
    x = torch.randn(3, 32)
    y = lin(x)
    ref = x @ base.t()
    # int8 groupwise base: ~1% relative error
    assert (y - ref).abs().max() / ref.abs().max() < 0.02
    # trains through LoRA
    y.sum().backward()
    assert lin.lora_a.grad is not None and lin.lora_b.grad is not None
    assert not any(p.requires_grad for n, p in lin.named_parameters()
                   if "lora" not in n)


def _sparse_worker(rank, world):
    import deepspeed_amd
    torch.manual_seed(2)
    emb = torch.nn.EmbeddingBag(20, 8, mode="mean", sparse=True)
    engine, _, _, _ = deepspeed_amd.initialize(
        model=emb, config={"train_micro_batch_size_per_gpu": 1,
                           "optimizer": {"type": "sgd",
                                         "params": {"lr": 0.1}}})
    ids = torch.tensor([[rank, rank + 5]])  # different rows per rank
    out = engine(ids)
    loss = out.sum()
    engine.backward(loss)
    engine.step()
    # all ranks end with identical weights (sparse grads were averaged)
    import torch.distributed as td
    w = emb.weight.detach().to_dense() if emb.weight.is_sparse \
        else emb.weight.detach()
    peers = [torch.empty_like(w) for _ in range(world)]
    td.all_gather(peers, w)
    assert torch.equal(peers[0], peers[1])


def test_sparse_grad_allreduce():
    run_distributed(_sparse_worker, world_size=2)


def test_ds_report_runs(capsys):
    from deepspeed_amd.utils.ds_report import main
    assert main() == 0
    out = capsys.readouterr().out
    assert "op availability" in out and "fused_adam_flat" in out
