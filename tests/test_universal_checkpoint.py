"""Universal checkpoint tests (reference contract:
tests/unit/checkpoint/test_universal_checkpoint.py): save at DP=2, convert
offline, resume at DP=1 with identical optimizer math.
"""

import os

import torch

from .common import run_distributed, run_local

_CONFIG = {
    "train_micro_batch_size_per_gpu": 2,
    "zero_optimization": {"stage": 1, "overlap_comm": False},
    "optimizer": {"type": "AdamW",
                  "params": {"lr": 1e-3, "weight_decay": 0.01}},
}


def _build_engine(stage=1):
    import deepspeed_amd
    from deepspeed_amd.models import LlamaForCausalLM, llama_tiny
    import copy
    torch.manual_seed(31)
    model = LlamaForCausalLM(llama_tiny())
    cfg = copy.deepcopy(_CONFIG)
    cfg["zero_optimization"]["stage"] = stage
    engine, opt, _, _ = deepspeed_amd.initialize(model=model, config=cfg)
    return engine, opt


def _same_batches(n):
    g = torch.Generator().manual_seed(77)
    return [torch.randint(0, 512, (2, 32), generator=g) for _ in range(n)]


def _train(engine, batches):
    for ids in batches:
        loss = engine(ids, labels=ids)
        engine.backward(loss)
        engine.step()
    return loss


def _save_worker(rank, world, tmp, stage=1):
    engine, _ = _build_engine(stage)
    _train(engine, _same_batches(2))  # identical batch on both ranks
    engine.save_checkpoint(tmp, tag="step2")
    # continue one more step; dump resulting fp32 masters for comparison
    _train(engine, _same_batches(3)[2:])
    if hasattr(engine.optimizer, "get_fp32_state_dict"):
        fp32 = engine.optimizer.get_fp32_state_dict(engine.module)
    else:
        fp32 = engine.optimizer.get_full_state_dict(dtype=torch.float32)
    if rank == 0:
        torch.save(fp32, os.path.join(tmp, "ref_after3.pt"))


def _resume_worker(rank, world, tmp, stage=1):
    engine, opt = _build_engine(stage)
    engine.load_checkpoint(tmp, tag="step2", load_universal=True)
    _train(engine, _same_batches(3)[2:])
    if hasattr(opt, "get_fp32_state_dict"):
        fp32 = opt.get_fp32_state_dict(engine.module)
    else:
        fp32 = opt.get_full_state_dict(dtype=torch.float32)
    if rank != 0:
        return  # consolidated fp32 export lands on rank 0 only
    ref = torch.load(os.path.join(tmp, "ref_after3.pt"), weights_only=False)
    assert set(fp32) == set(ref)
    for k in ref:
        torch.testing.assert_close(fp32[k], ref[k], rtol=1e-5, atol=1e-6), k


def test_universal_checkpoint_dp2_to_dp1(tmp_path):
    tmp = str(tmp_path)
    run_distributed(_save_worker, world_size=2, args=(tmp,))

    from deepspeed_amd.checkpoint import ds_to_universal
    out = ds_to_universal(tmp, tag="step2")
    usd = torch.load(os.path.join(out, "universal_optim_states.pt"),
                     weights_only=False)
    assert usd["step"] == 2
    assert "model.layers.0.self_attn.q_proj.weight" in usd["param"]
    # universal params must equal the gathered fp32 truth at save time
    assert all(v.dtype == torch.float32 for v in usd["param"].values())

    run_local(_resume_worker, args=(tmp,))


def test_universal_checkpoint_zero3_dp2_to_dp1(tmp_path):
    tmp = str(tmp_path)
    run_distributed(_save_worker, world_size=2, args=(tmp, 3))
    from deepspeed_amd.checkpoint import ds_to_universal
    ds_to_universal(tmp, tag="step2")
    run_local(_resume_worker, args=(tmp, 3))


def test_universal_checkpoint_dp4_to_dp2(tmp_path):
    """Elastic resume shrinking 4 -> 2 ranks (not just to a single rank)."""
    tmp = str(tmp_path)
    run_distributed(_save_worker, world_size=4, args=(tmp, 1))
    from deepspeed_amd.checkpoint import ds_to_universal
    ds_to_universal(tmp, tag="step2")
    run_distributed(_resume_worker, world_size=2, args=(tmp, 1))


import pytest


@pytest.mark.parametrize("stage", [1, 2, 3])
@pytest.mark.parametrize("save_ws,load_ws", [(4, 2), (2, 4)])
def test_universal_reshape_matrix(tmp_path, stage, save_ws, load_ws):
    """Cross-world-size reshaping matrix, shrink AND grow, for every ZeRO
    stage including optimizer state (reference DistributedFixture pattern,
    tests/unit/common.py:354 + tests/unit/checkpoint/)."""
    tmp = str(tmp_path)
    run_distributed(_save_worker, world_size=save_ws, args=(tmp, stage))
    from deepspeed_amd.checkpoint import ds_to_universal
    ds_to_universal(tmp, tag="step2")
    run_distributed(_resume_worker, world_size=load_ws, args=(tmp, stage))


def _moe_uni_save_worker(rank, world, tmp):
    import deepspeed_amd
    from deepspeed_amd.models import MixtralForCausalLM, mixtral_tiny

    torch.manual_seed(17)
    cfg = mixtral_tiny(ep_size=2, num_experts=4)
    model = MixtralForCausalLM(cfg)
    engine, _, _, _ = deepspeed_amd.initialize(model=model, config={
        "train_micro_batch_size_per_gpu": 2,
        "zero_optimization": {"stage": 1, "overlap_comm": False},
        "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}}})
    g = torch.Generator().manual_seed(70)
    for _ in range(2):
        ids = torch.randint(0, cfg.vocab_size, (2, 16), generator=g)
        loss = engine(ids, labels=ids)
        engine.backward(loss)
        engine.step()
    engine.save_checkpoint(tmp, tag="moe0")
    # dump this rank's expert params for comparison, keyed by ep rank
    from deepspeed_amd.parallel import groups
    name = next(m.expert_group_name for m in engine.module.modules()
                if hasattr(m, "expert_group_name"))
    ep_rank = groups.get_expert_parallel_rank(name)
    if torch.distributed.get_rank(
            groups.get_expert_data_parallel_group(name)) == 0:
        experts = {n: p.detach().clone()
                   for n, p in engine.module.named_parameters()
                   if getattr(p, "allreduce", True) is False}
        torch.save(experts, os.path.join(tmp, f"want_ep{ep_rank}.pt"))


def _moe_uni_resume_worker(rank, world, tmp):
    import deepspeed_amd
    from deepspeed_amd.models import MixtralForCausalLM, mixtral_tiny

    torch.manual_seed(99)  # different init: everything must come from ckpt
    cfg = mixtral_tiny(ep_size=2, num_experts=4)
    model = MixtralForCausalLM(cfg)
    engine, _, _, _ = deepspeed_amd.initialize(model=model, config={
        "train_micro_batch_size_per_gpu": 2,
        "zero_optimization": {"stage": 1, "overlap_comm": False},
        "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}}})
    engine.load_checkpoint(tmp, tag="moe0", load_universal=True)
    from deepspeed_amd.parallel import groups
    name = next(m.expert_group_name for m in engine.module.modules()
                if hasattr(m, "expert_group_name"))
    ep_rank = groups.get_expert_parallel_rank(name)
    want = torch.load(os.path.join(tmp, f"want_ep{ep_rank}.pt"),
                      weights_only=False)
    for n, p in engine.module.named_parameters():
        if getattr(p, "allreduce", True) is False:
            torch.testing.assert_close(p.detach(), want[n], rtol=1e-5,
                                       atol=1e-6)


def test_universal_moe_expert_dp_reshape(tmp_path):
    """MoE universal resume: EP=2 fixed, expert-DP reshaped 2 -> 1
    (world 4 -> 2). Expert params are keyed per EP rank in the universal
    files (name@epR) so same-named experts from different EP ranks do not
    collide."""
    tmp = str(tmp_path)
    run_distributed(_moe_uni_save_worker, world_size=4, args=(tmp,))
    from deepspeed_amd.checkpoint import ds_to_universal
    ds_to_universal(tmp, tag="moe0")
    run_distributed(_moe_uni_resume_worker, world_size=2, args=(tmp,))


def test_universal_cross_stage_z2_to_z3(tmp_path):
    """Save under ZeRO-2, convert, resume under ZeRO-3 (same DP): the
    universal layout is stage-agnostic (per-param fp32 + moments)."""
    tmp = str(tmp_path)
    run_distributed(_save_worker, world_size=2, args=(tmp, 2))
    from deepspeed_amd.checkpoint import ds_to_universal
    ds_to_universal(tmp, tag="step2")
    run_distributed(_resume_worker, world_size=2, args=(tmp, 3))


def test_universal_cross_stage_z3_to_z1(tmp_path):
    tmp = str(tmp_path)
    run_distributed(_save_worker, world_size=2, args=(tmp, 3))
    from deepspeed_amd.checkpoint import ds_to_universal
    ds_to_universal(tmp, tag="step2")
    run_distributed(_resume_worker, world_size=2, args=(tmp, 1))
