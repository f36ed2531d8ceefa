"""zero.Init construction-time semantics, including the meta-device mode
that lets models larger than device+host memory be constructed and trained
(reference contract: tests/unit/runtime/zero/test_zero_context*.py and
partition_parameters.py Init :824 / remote_device handling)."""

import pytest
import torch
import torch.nn as nn

from .common import run_distributed


def test_init_casts_registered_params():
    from deepspeed_amd.runtime.zero.partition import Init
    with Init(dtype=torch.bfloat16):
        m = nn.Linear(8, 8)
    assert m.weight.dtype == torch.bfloat16
    assert m.bias.dtype == torch.bfloat16


def test_init_nested_contexts():
    from deepspeed_amd.runtime.zero.partition import Init
    with Init(dtype=torch.bfloat16):
        with Init(dtype=torch.bfloat16):
            inner = nn.Linear(4, 4)
        outer = nn.Linear(4, 4)
    # both constructed under a live context; unpatching must be clean
    m = nn.Linear(4, 4)
    assert inner.weight.dtype == torch.bfloat16
    assert outer.weight.dtype == torch.bfloat16
    assert m.weight.dtype == torch.float32


def test_init_meta_allocates_nothing():
    from deepspeed_amd.runtime.zero.partition import Init
    with Init(remote_device="meta"):
        m = nn.Linear(64, 64)
    assert m.weight.is_meta and m.bias.is_meta
    assert m.weight.dtype == torch.bfloat16


def _meta_train_worker(rank, world):
    import deepspeed_amd
    from deepspeed_amd import comm as dist
    from deepspeed_amd.models import LlamaForCausalLM, llama_tiny
    from deepspeed_amd.runtime.zero.partition import Init

    cfg = llama_tiny()
    with Init(remote_device="meta"):
        model = LlamaForCausalLM(cfg)
    assert all(p.is_meta for p in model.parameters())

    engine, opt, _, _ = deepspeed_amd.initialize(model=model, config={
        "train_micro_batch_size_per_gpu": 2,
        "zero_optimization": {"stage": 3, "overlap_comm": False},
        "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}}})

    # materialized + sharded: no meta params remain, loss is finite and
    # decreases over a few steps (weights were reset_parameters()-drawn)
    torch.manual_seed(7 + rank)
    losses = []
    for _ in range(5):
        ids = torch.randint(0, cfg.vocab_size, (2, 32))
        loss = engine(ids, labels=ids)
        engine.backward(loss)
        engine.step()
        losses.append(loss.item())
    assert all(torch.isfinite(torch.tensor(losses)))
    assert losses[-1] < losses[0], losses

    # rank-0 broadcast made every rank's shards consistent: gathering a
    # unit on each rank must produce identical full weights
    opt3 = engine.optimizer
    u = opt3.units[0]
    with opt3.gathered_params([u.params[0]]):
        t = u.params[0].data.clone().float()
    ref = t.clone()
    dist.broadcast(ref, src=0)
    assert torch.equal(ref, t)
    # and the rank-0 full state dict exports every parameter, no metas
    sd = opt3.get_full_state_dict()
    if rank == 0:
        names = {n for n, _ in engine.module.named_parameters()}
        assert set(sd.keys()) == names
        assert not any(v.is_meta for v in sd.values())


def test_zero3_meta_init_train_ws2():
    run_distributed(_meta_train_worker, world_size=2)


def _meta_resume_worker(rank, world, tmp):
    """meta-init + load_checkpoint: the no-full-weights restore path."""
    import deepspeed_amd
    from deepspeed_amd.models import LlamaForCausalLM, llama_tiny
    from deepspeed_amd.runtime.zero.partition import Init

    cfg = llama_tiny()

    def make(meta):
        if meta:
            with Init(remote_device="meta"):
                m = LlamaForCausalLM(cfg)
        else:
            torch.manual_seed(11)
            m = LlamaForCausalLM(cfg)
        engine, _, _, _ = deepspeed_amd.initialize(model=m, config={
            "train_micro_batch_size_per_gpu": 2,
            "bf16": {"enabled": True},
            "zero_optimization": {"stage": 3, "overlap_comm": False},
            "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}}})
        return engine

    e1 = make(meta=False)
    torch.manual_seed(3)
    for _ in range(2):
        ids = torch.randint(0, cfg.vocab_size, (2, 16))
        loss = e1.forward(ids, labels=ids)
        e1.backward(loss)
        e1.step()
    e1.save_checkpoint(tmp, tag="t0")
    sd1 = e1.optimizer.get_full_state_dict()

    e2 = make(meta=True)
    e2.load_checkpoint(tmp, tag="t0")
    sd2 = e2.optimizer.get_full_state_dict()
    if rank == 0:
        for k in sd1:
            assert torch.equal(sd1[k], sd2[k]), k


def test_zero3_meta_init_resume_ws2(tmp_path):
    run_distributed(_meta_resume_worker, world_size=2, args=(str(tmp_path),))


def test_meta_init_full_capacity_pipeline_ws2():
    """The 70B-class capacity pipeline end to end at mini scale:
    zero.Init(remote_device='meta') construction (no weight allocation)
    -> ZeRO-3 unit-wise materialization -> offload_param host shards ->
    offload_optimizer host Adam. Parity vs the plain device pipeline."""
    run_distributed(_capacity_worker, world_size=2)


def _capacity_worker(rank, world):
    import copy
    import deepspeed_amd
    from deepspeed_amd import zero
    from deepspeed_amd.models import LlamaForCausalLM, llama_tiny

    def cfg(capacity):
        z = {"stage": 3, "overlap_comm": False,
             "stage3_param_persistence_threshold": 64}
        if capacity:
            z["offload_param"] = {"device": "cpu", "pin_memory": True}
            z["offload_optimizer"] = {"device": "cpu", "pin_memory": True}
        return {"train_micro_batch_size_per_gpu": 2,
                "bf16": {"enabled": True},
                "zero_optimization": z,
                "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}}}

    def data(n):
        torch.manual_seed(5)
        return [torch.randint(0, 500, (2, 32)) for _ in range(n)]

    def run(capacity):
        if capacity:
            with zero.Init(remote_device="meta"):
                model = LlamaForCausalLM(llama_tiny())
            assert all(p.is_meta or p.numel() == 0
                       for p in model.parameters())
        else:
            torch.manual_seed(31)
            model = LlamaForCausalLM(llama_tiny())
        engine, _, _, _ = deepspeed_amd.initialize(model=model,
                                                   config=cfg(capacity))
        losses = []
        for ids in data(3):
            loss = engine(ids.to(engine.device), labels=ids.to(engine.device))
            engine.backward(loss)
            engine.step()
            losses.append(loss.item())
        return losses

    base = run(False)
    cap = run(True)
    # meta materialization draws its own init (reset_parameters on rank 0,
    # broadcast) so trajectories differ numerically — both must train
    for ls in (base, cap):
        assert all(torch.isfinite(torch.tensor(ls))), ls
        assert ls[-1] < ls[0] + 0.5, ls
