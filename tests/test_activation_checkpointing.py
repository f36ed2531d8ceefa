"""Activation checkpointing must reproduce the non-checkpointed loss and
gradients exactly (reference contract:
tests/unit/runtime/activation_checkpointing/test_activation_checkpointing.py).
"""

import copy

import torch

from deepspeed_amd.models import LlamaForCausalLM, llama_tiny


def _loss_and_grads(model, ids):
    model.zero_grad(set_to_none=True)
    loss = model(ids, labels=ids)
    loss.backward()
    return loss.detach().clone(), [p.grad.clone() for p in model.parameters()]


def test_checkpoint_grad_parity():
    torch.manual_seed(5)
    model = LlamaForCausalLM(llama_tiny())
    model.train()
    ids = torch.randint(0, 512, (2, 64))

    loss_ref, grads_ref = _loss_and_grads(model, ids)
    model.model.gradient_checkpointing_enable()
    loss_ckpt, grads_ckpt = _loss_and_grads(model, ids)

    torch.testing.assert_close(loss_ckpt, loss_ref, rtol=1e-6, atol=1e-6)
    for g1, g2 in zip(grads_ckpt, grads_ref):
        torch.testing.assert_close(g1, g2, rtol=1e-5, atol=1e-6)


def test_checkpoint_rng_replay():
    """Dropout inside a checkpointed function must replay the forward mask."""
    from deepspeed_amd.runtime.activation_checkpointing import checkpoint

    torch.manual_seed(0)
    lin = torch.nn.Linear(16, 16)

    def fn(x):
        return torch.nn.functional.dropout(lin(x), p=0.5, training=True)

    x = torch.randn(4, 16, requires_grad=True)
    torch.manual_seed(123)
    out = checkpoint(fn, x)
    # plain run with the same seed gives the same mask
    torch.manual_seed(123)
    ref = fn(x)
    torch.testing.assert_close(out, ref)
    # backward recompute uses the saved RNG state, so grads match the ref graph
    g = torch.randn_like(out)
    out.backward(g)
    x_grad = x.grad.clone()
    x.grad = None
    ref.backward(g)
    torch.testing.assert_close(x_grad, x.grad)


def _ckpt_zero2_worker(rank, world):
    import deepspeed_amd

    def make():
        torch.manual_seed(9)
        return LlamaForCausalLM(llama_tiny())

    config = {
        "train_micro_batch_size_per_gpu": 2,
        "zero_optimization": {"stage": 2, "overlap_comm": False},
        "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}},
    }
    results = []
    for use_ckpt in (False, True):
        model = make()
        if use_ckpt:
            model.model.gradient_checkpointing_enable()
        engine, _, _, _ = deepspeed_amd.initialize(
            model=model, config=copy.deepcopy(config))
        torch.manual_seed(3)
        for _ in range(3):
            ids = torch.randint(0, 512, (2, 32))
            loss = engine(ids, labels=ids)
            engine.backward(loss)
            engine.step()
        results.append([p.detach().clone() for p in engine.module.parameters()])
    for a, b in zip(results[0], results[1]):
        torch.testing.assert_close(a, b, rtol=1e-5, atol=1e-6)


def test_checkpoint_with_zero2_engine():
    """Engine-level: checkpointed training matches non-checkpointed weights.

    Runs in a child process (run_local): initializing gloo in the pytest main
    process would deadlock later fork-based multi-process tests."""
    from .common import run_local
    run_local(_ckpt_zero2_worker)
