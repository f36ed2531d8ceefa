"""Pipeline-parallel tests (reference contract:
tests/unit/runtime/pipe/test_pipe.py + test_topology.py): layer
partitioning, 1F1B schedule shape, and exact loss/weight parity of a
2-stage pipeline against the same model run sequentially in one process.
"""

import os

import torch
import torch.nn as nn

from .common import run_distributed


def test_partition_balanced():
    from deepspeed_amd.runtime.pipe.module import partition_balanced
    assert partition_balanced([1, 1, 1, 1], 2) == [0, 2, 4]
    assert partition_balanced([5, 1, 1, 1], 2) == [0, 1, 4]
    b = partition_balanced([3, 3, 3, 3, 3, 3, 3, 3], 4)
    assert b == [0, 2, 4, 6, 8]
    # every part non-empty even with skewed weights
    b = partition_balanced([100, 1, 1, 1], 4)
    assert b == [0, 1, 2, 3, 4]


def test_train_schedule_1f1b():
    from deepspeed_amd.runtime.pipe import schedule as s
    M, S = 4, 2
    for stage in range(S):
        sch = s.TrainSchedule(M, S, stage)
        cmds = [c for step in sch for c in step]
        fwd = [c for c in cmds if isinstance(c, s.ForwardPass)]
        bwd = [c for c in cmds if isinstance(c, s.BackwardPass)]
        assert len(fwd) == M and len(bwd) == M
        # every backward of mb i comes after its forward
        order = {}
        for i, c in enumerate(cmds):
            if isinstance(c, (s.ForwardPass, s.BackwardPass)):
                order[(type(c).__name__, c.micro_batch_id)] = i
        for mb in range(M):
            assert order[("ForwardPass", mb)] < order[("BackwardPass", mb)]
        # finishes with reduce + step
        assert isinstance(cmds[-1], s.OptimizerStep)
        assert isinstance(cmds[-2], s.ReduceGrads)
        # stage 0 holds at most warmup+1 = S-stage buffers
        assert sch.num_pipe_buffers() == min(S - stage - 1, M) + 1


def _make_layers(seed=11):
    torch.manual_seed(seed)
    return [nn.Linear(16, 32), nn.ReLU(), nn.Linear(32, 32), nn.ReLU(),
            nn.Linear(32, 4)]


def _make_data(n_batches, micro_bs, seed=123):
    g = torch.Generator().manual_seed(seed)
    return [(torch.randn(micro_bs, 16, generator=g),
             torch.randn(micro_bs, 4, generator=g)) for _ in range(n_batches)]


_CONFIG = {
    "train_micro_batch_size_per_gpu": 4,
    "gradient_accumulation_steps": 2,
    "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}},
}


def _pipe_parity_worker(rank, world):
    import deepspeed_amd
    from deepspeed_amd.runtime.pipe.module import PipelineModule

    loss_fn = nn.functional.mse_loss
    net = PipelineModule(_make_layers(), num_stages=world, loss_fn=loss_fn,
                         partition_method="parameters")
    engine, _, _, _ = deepspeed_amd.initialize(model=net, config=dict(_CONFIG))

    n_steps, mbs = 3, _CONFIG["gradient_accumulation_steps"]
    data = _make_data(n_steps * mbs, 4)
    it = iter(data)
    pipe_losses = [engine.train_batch(it).item() for _ in range(n_steps)]

    # sequential reference (same init, same data, same optimizer math)
    ref = nn.Sequential(*_make_layers())
    opt = torch.optim.AdamW(ref.parameters(), lr=1e-3)
    ref_losses = []
    di = iter(data)
    for _ in range(n_steps):
        tot = 0.0
        for _ in range(mbs):
            x, y = next(di)
            loss = loss_fn(ref(x), y)
            (loss / mbs).backward()
            tot += loss.item()
        opt.step()
        opt.zero_grad()
        ref_losses.append(tot / mbs)

    for got, want in zip(pipe_losses, ref_losses):
        assert abs(got - want) < 1e-5, (pipe_losses, ref_losses)

    # my stage's params must equal the reference slice
    ref_layers = list(ref)
    mine = net.forward_funcs
    ref_slice = ref_layers[net.part_start:net.part_end]
    for m, r in zip(mine, ref_slice):
        if isinstance(m, nn.Module):
            for pm, pr in zip(m.parameters(), r.parameters()):
                # FusedAdam's update order differs from torch.AdamW at the
                # last-ulp level; parity is at fp32 noise scale
                torch.testing.assert_close(pm, pr, rtol=1e-4, atol=2e-5)


def test_pipeline_2stage_parity():
    run_distributed(_pipe_parity_worker, world_size=2)


def _tied_worker(rank, world):
    import deepspeed_amd
    from deepspeed_amd.runtime.pipe.module import (PipelineModule,
                                                   TiedLayerSpec, LayerSpec)
    import torch.distributed as td

    specs = [
        TiedLayerSpec("emb", nn.Linear, 8, 8),
        LayerSpec(nn.Linear, 8, 8),
        LayerSpec(nn.Linear, 8, 8),
        TiedLayerSpec("emb", nn.Linear, 8, 8),
    ]
    net = PipelineModule(specs, num_stages=world, loss_fn=nn.functional.mse_loss,
                         partition_method="uniform")
    engine, _, _, _ = deepspeed_amd.initialize(model=net, config=dict(_CONFIG))
    g = torch.Generator().manual_seed(7)
    data = [(torch.randn(4, 8, generator=g), torch.randn(4, 8, generator=g))
            for _ in range(8)]
    it = iter(data)
    for _ in range(2):
        loss = engine.train_batch(it)
        assert torch.isfinite(loss)

    # tied copies must be identical across the stages that hold them
    w = net.tied_modules["emb"].weight.detach().clone()
    peers = [torch.empty_like(w) for _ in range(world)]
    td.all_gather(peers, w)
    for p in peers:
        torch.testing.assert_close(p, peers[0])


def test_pipeline_tied_weights():
    run_distributed(_tied_worker, world_size=2)


def test_pipeline_4stage_parity():
    run_distributed(_pipe_parity_worker, world_size=4)


def _pp2dp2_worker(rank, world):
    """Hybrid PP=2 x DP=2: each stage has two data-parallel replicas; the
    per-stage DP all-reduce must average their gradients."""
    import deepspeed_amd
    from deepspeed_amd.runtime.pipe.module import PipelineModule

    loss_fn = nn.functional.mse_loss
    net = PipelineModule(_make_layers(), num_stages=2, loss_fn=loss_fn,
                         partition_method="parameters")
    assert net.grid.data_parallel_size == 2
    engine, _, _, _ = deepspeed_amd.initialize(model=net, config=dict(_CONFIG))

    n_steps, mbs = 2, _CONFIG["gradient_accumulation_steps"]
    dp = net.grid.data_parallel_id
    data_all = [_make_data(n_steps * mbs, 4, seed=123 + d) for d in range(2)]
    it = iter(data_all[dp])
    losses = [engine.train_batch(it).item() for _ in range(n_steps)]

    # reference: grads averaged over BOTH dp streams
    ref = nn.Sequential(*_make_layers())
    opt = torch.optim.AdamW(ref.parameters(), lr=1e-3)
    its = [iter(data_all[d]) for d in range(2)]
    ref_losses = []
    for _ in range(n_steps):
        tot = 0.0
        for d in range(2):
            for _ in range(mbs):
                x, y = next(its[d])
                loss = loss_fn(ref(x), y)
                (loss / (mbs * 2)).backward()
                tot += loss.item()
        opt.step()
        opt.zero_grad()
        ref_losses.append(tot / (mbs * 2))

    for got, want in zip(losses, ref_losses):
        assert abs(got - want) < 1e-5, (losses, ref_losses)
    ref_slice = list(ref)[net.part_start:net.part_end]
    for m, r in zip(net.forward_funcs, ref_slice):
        if isinstance(m, nn.Module):
            for pm, pr in zip(m.parameters(), r.parameters()):
                torch.testing.assert_close(pm, pr, rtol=1e-4, atol=2e-5)


def test_pipeline_pp2_dp2_hybrid():
    run_distributed(_pp2dp2_worker, world_size=4)


def _pp_resume_worker(rank, world, ckpt_dir):
    """PP=2 x DP=2 checkpoint save/resume: stage-qualified checkpoint
    files (each stage owns different layers — unqualified names would
    collide), and a resumed engine must continue exactly like the
    original."""
    import deepspeed_amd
    from deepspeed_amd.runtime.pipe.module import PipelineModule

    loss_fn = nn.functional.mse_loss

    def build():
        net = PipelineModule(_make_layers(), num_stages=2, loss_fn=loss_fn,
                             partition_method="parameters")
        eng, _, _, _ = deepspeed_amd.initialize(model=net,
                                                config=dict(_CONFIG))
        return net, eng

    net, engine = build()
    mbs = _CONFIG["gradient_accumulation_steps"]
    dp = net.grid.data_parallel_id
    data = _make_data(6 * mbs, 4, seed=321 + dp)
    it = iter(data)
    for _ in range(2):
        engine.train_batch(it)
    engine.save_checkpoint(ckpt_dir)

    # distinct stages must have produced distinct model-state files
    if rank == 0:
        import glob, os
        tag_dir = os.path.join(
            ckpt_dir, open(os.path.join(ckpt_dir, "latest")).read().strip())
        files = glob.glob(os.path.join(tag_dir, "*pp_rank*_model_states.pt"))
        assert len(files) == 2, files
    cont = engine.train_batch(it).item()

    net2, engine2 = build()
    load_tag, _ = engine2.load_checkpoint(ckpt_dir)
    assert load_tag is not None
    assert engine2.global_steps == engine.global_steps - 1
    it2 = iter(data)
    for _ in range(2 * mbs):   # consume the pre-checkpoint batches
        next(it2)
    resumed = engine2.train_batch(it2).item()
    assert abs(resumed - cont) < 1e-6, (resumed, cont)


def test_pipeline_checkpoint_resume_pp2dp2(tmp_path):
    run_distributed(_pp_resume_worker, world_size=4, args=(str(tmp_path),))


def _pp2_universal_save_worker(rank, world, tmp):
    import deepspeed_amd
    from deepspeed_amd.runtime.pipe.module import PipelineModule
    cfg = dict(_CONFIG)
    cfg["zero_optimization"] = {"stage": 1, "overlap_comm": False}
    net = PipelineModule(_make_layers(), num_stages=2,
                         loss_fn=nn.functional.mse_loss,
                         partition_method="parameters")
    engine, _, _, _ = deepspeed_amd.initialize(model=net, config=cfg)
    mbs = _CONFIG["gradient_accumulation_steps"]
    dp = net.grid.data_parallel_id
    it = iter(_make_data(3 * mbs, 4, seed=555 + dp))
    for _ in range(2):
        engine.train_batch(it)
    engine.save_checkpoint(tmp, tag="pp2")
    cont = engine.train_batch(it).item()
    if rank == 0:
        import json
        with open(os.path.join(tmp, "cont.json"), "w") as f:
            json.dump({"cont": cont}, f)


def _pp1_universal_resume_worker(rank, world, tmp):
    import deepspeed_amd
    from deepspeed_amd.runtime.pipe.module import PipelineModule
    cfg = dict(_CONFIG)
    cfg["zero_optimization"] = {"stage": 1, "overlap_comm": False}
    net = PipelineModule(_make_layers(), num_stages=1,
                         loss_fn=nn.functional.mse_loss,
                         partition_method="parameters")
    engine, _, _, _ = deepspeed_amd.initialize(model=net, config=cfg)
    tag_dir, _ = engine.load_checkpoint(tmp, tag="pp2", load_universal=True)
    assert tag_dir is not None
    mbs = _CONFIG["gradient_accumulation_steps"]
    dp = net.grid.data_parallel_id
    it = iter(_make_data(3 * mbs, 4, seed=555 + dp))
    for _ in range(2 * mbs):
        next(it)
    resumed = engine.train_batch(it).item()
    import json
    with open(os.path.join(tmp, "cont.json")) as f:
        cont = json.load(f)["cont"]
    assert abs(resumed - cont) < 1e-5, (resumed, cont)


def test_pipeline_degree_reshape_pp2_to_pp1(tmp_path):
    """Save at PP=2 x DP=2, convert the ZeRO state to universal, resume at
    PP=1 x DP=2: per-global-layer module files + name-keyed universal
    optimizer state make the pipeline degree a resume-time choice."""
    tmp = str(tmp_path)
    run_distributed(_pp2_universal_save_worker, world_size=4, args=(tmp,))
    from deepspeed_amd.checkpoint import ds_to_universal
    ds_to_universal(tmp, tag="pp2")
    run_distributed(_pp1_universal_resume_worker, world_size=2, args=(tmp,))


def _tied_reshape_worker_save(rank, world, tmp):
    """Save a PP2 pipeline with tied first/last layers; the tied module
    file is written once by the lowest owning stage."""
    import deepspeed_amd
    from deepspeed_amd.runtime.pipe.module import (PipelineModule,
                                                   TiedLayerSpec, LayerSpec)
    cfg = dict(_CONFIG)
    cfg["zero_optimization"] = {"stage": 1, "overlap_comm": False}
    specs = [TiedLayerSpec("emb", nn.Linear, 8, 8),
             LayerSpec(nn.Linear, 8, 8),
             LayerSpec(nn.Linear, 8, 8),
             TiedLayerSpec("emb", nn.Linear, 8, 8)]
    net = PipelineModule(specs, num_stages=2,
                         loss_fn=nn.functional.mse_loss,
                         partition_method="uniform")
    engine, _, _, _ = deepspeed_amd.initialize(model=net, config=cfg)
    g = torch.Generator().manual_seed(7)
    data = [(torch.randn(4, 8, generator=g),
             torch.randn(4, 8, generator=g)) for _ in range(6)]
    it = iter(data)
    for _ in range(2):
        engine.train_batch(it)
    engine.save_checkpoint(tmp, tag="tied")
    if rank == 0:
        torch.save(net.tied_modules["emb"].weight.detach().clone(),
                   os.path.join(tmp, "tied_w.pt"))


def _tied_reshape_worker_load(rank, world, tmp):
    import deepspeed_amd
    from deepspeed_amd.runtime.pipe.module import (PipelineModule,
                                                   TiedLayerSpec, LayerSpec)
    cfg = dict(_CONFIG)
    cfg["zero_optimization"] = {"stage": 1, "overlap_comm": False}
    specs = [TiedLayerSpec("emb", nn.Linear, 8, 8),
             LayerSpec(nn.Linear, 8, 8),
             LayerSpec(nn.Linear, 8, 8),
             TiedLayerSpec("emb", nn.Linear, 8, 8)]
    net = PipelineModule(specs, num_stages=1,
                         loss_fn=nn.functional.mse_loss,
                         partition_method="uniform")
    engine, _, _, _ = deepspeed_amd.initialize(model=net, config=cfg)
    tag, _ = engine.load_checkpoint(tmp, tag="tied", load_universal=True)
    assert tag is not None
    want = torch.load(os.path.join(tmp, "tied_w.pt"), weights_only=True)
    torch.testing.assert_close(net.tied_modules["emb"].weight.detach(),
                               want)


def test_pipeline_tied_reshape_pp2_to_pp1(tmp_path):
    """Cross-PP-degree resume with TIED layers: the per-key tied file
    restores the single shared module at PP=1."""
    tmp = str(tmp_path)
    run_distributed(_tied_reshape_worker_save, world_size=2, args=(tmp,))
    from deepspeed_amd.checkpoint import ds_to_universal
    ds_to_universal(tmp, tag="tied")
    run_distributed(_tied_reshape_worker_load, world_size=1, args=(tmp,))
