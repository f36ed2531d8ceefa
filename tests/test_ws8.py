"""World-size-8 (gloo) hardening of every multi-rank path the driver's
8-GPU scaling bench exercises: ZeRO-1/2/3, hpZ, MoE EP=8, Ulysses SP=8,
PP4xDP2, TP2xDP4, and a ZeRO-3 overlap/backpressure stress with tight
prefetch budgets (reference contract: tests/unit/runtime/zero/test_zero.py
at world_size 4; here widened to the full single-node rank count)."""

import copy

import pytest
import torch
import torch.nn as nn

from .common import run_distributed
from .test_zero12 import _zero_worker


@pytest.mark.parametrize("stage", [1, 2, 3])
def test_zero_parity_ws8(stage):
    run_distributed(_zero_worker, world_size=8,
                    args=(stage, 1, 0.0, "bf16"), timeout=600)


def test_zero3_hpz_ws8():
    # hierarchical secondary shards with 2 sub-groups of 4
    run_distributed(_zero_worker, world_size=8,
                    args=(3, 1, 0.0, "bf16", 4), timeout=600)


def _mixtral_ep8_worker(rank, world):
    import deepspeed_amd
    from deepspeed_amd import comm as dist
    from deepspeed_amd.models import MixtralForCausalLM, mixtral_tiny

    torch.manual_seed(17)
    cfg = mixtral_tiny(ep_size=8, num_experts=8)
    model = MixtralForCausalLM(cfg)
    engine, opt, _, _ = deepspeed_amd.initialize(model=model, config={
        "train_micro_batch_size_per_gpu": 2,
        "zero_optimization": {"stage": 2, "overlap_comm": True},
        "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}}})
    torch.manual_seed(50 + rank)
    for _ in range(2):
        ids = torch.randint(0, cfg.vocab_size, (2, 32))
        loss = engine(ids, labels=ids)
        engine.backward(loss)
        engine.step()
        assert torch.isfinite(loss)
    for p in engine.module.parameters():
        if getattr(p, "allreduce", True) is False:
            continue
        ref = p.data.clone()
        dist.broadcast(ref, src=0)
        assert torch.equal(ref, p.data), "dense param diverged across ranks"


def test_mixtral_ep8_ws8():
    run_distributed(_mixtral_ep8_worker, world_size=8, timeout=600)


def _ulysses_sp8_worker(rank, world):
    import deepspeed_amd
    from deepspeed_amd.models import LlamaForCausalLM
    from deepspeed_amd.models.llama import LlamaConfig, enable_ulysses
    from deepspeed_amd.parallel import groups

    groups.initialize_sequence_parallel(world)
    cfg = LlamaConfig(vocab_size=256, hidden_size=64, intermediate_size=96,
                      num_layers=2, num_heads=8, num_kv_heads=8,
                      max_seq_len=64)
    torch.manual_seed(41)
    model = LlamaForCausalLM(cfg)
    enable_ulysses(model)
    engine, _, _, _ = deepspeed_amd.initialize(model=model, config={
        "train_micro_batch_size_per_gpu": 2,
        "zero_optimization": {"stage": 1, "overlap_comm": False},
        "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}}})

    torch.manual_seed(41)
    ref = LlamaForCausalLM(cfg)
    opt_ref = torch.optim.AdamW(ref.parameters(), lr=1e-3)

    torch.manual_seed(90)
    S = 64
    sl = slice(rank * S // world, (rank + 1) * S // world)
    for _ in range(2):
        ids = torch.randint(0, 256, (2, S + 1))
        x, y = ids[:, :-1], ids[:, 1:]
        pos = torch.arange(S, dtype=torch.int32).expand(2, S)
        loss = engine(x[:, sl].contiguous(), labels=y[:, sl].contiguous(),
                      positions=pos[:, sl].contiguous())
        engine.backward(loss)
        engine.step()
        l2 = ref(x, labels=y)
        l2.backward()
        opt_ref.step()
        opt_ref.zero_grad()
    for (n, p), (_, pr) in zip(engine.module.named_parameters(),
                               ref.named_parameters()):
        torch.testing.assert_close(p, pr, rtol=3e-4, atol=5e-4), n


def test_ulysses_sp8_ws8():
    run_distributed(_ulysses_sp8_worker, world_size=8, timeout=600)


def _pp4dp2_worker(rank, world):
    import deepspeed_amd
    from deepspeed_amd.runtime.pipe.module import PipelineModule

    def make_layers(seed=11):
        torch.manual_seed(seed)
        return [nn.Linear(16, 32), nn.Tanh(), nn.Linear(32, 32), nn.Tanh(),
                nn.Linear(32, 32), nn.Tanh(), nn.Linear(32, 4)]

    def make_data(n, bs, seed):
        g = torch.Generator().manual_seed(seed)
        return [(torch.randn(bs, 16, generator=g),
                 torch.randn(bs, 4, generator=g)) for _ in range(n)]

    cfg = {"train_micro_batch_size_per_gpu": 4,
           "gradient_accumulation_steps": 4,
           "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}}}
    loss_fn = nn.functional.mse_loss
    net = PipelineModule(make_layers(), num_stages=4, loss_fn=loss_fn,
                         partition_method="parameters")
    assert net.grid.data_parallel_size == 2
    engine, _, _, _ = deepspeed_amd.initialize(model=net, config=cfg)

    n_steps, mbs = 2, 4
    dp = net.grid.data_parallel_id
    data_all = [make_data(n_steps * mbs, 4, seed=123 + d) for d in range(2)]
    it = iter(data_all[dp])
    losses = [engine.train_batch(it).item() for _ in range(n_steps)]

    ref = nn.Sequential(*make_layers())
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


def test_pipeline_pp4_dp2_ws8():
    run_distributed(_pp4dp2_worker, world_size=8, timeout=600)


def _tp2dp4_worker(rank, world):
    import deepspeed_amd
    from deepspeed_amd.models import LlamaForCausalLM, llama_tiny
    from deepspeed_amd.parallel import groups

    torch.manual_seed(29)
    model = LlamaForCausalLM(llama_tiny())
    torch.manual_seed(29)
    ref = LlamaForCausalLM(llama_tiny())

    deepspeed_amd.tp_model_init(model, tp_size=2)
    assert groups.get_tensor_parallel_world_size() == 2
    assert groups.get_data_parallel_world_size() == 4
    engine, _, _, _ = deepspeed_amd.initialize(model=model, config={
        "train_micro_batch_size_per_gpu": 2,
        "zero_optimization": {"stage": 1, "overlap_comm": False},
        "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}}})
    opt_ref = torch.optim.AdamW(ref.parameters(), lr=1e-3)

    dp_rank = rank // 2  # tp pairs are contiguous
    torch.manual_seed(70)
    all_ids = [torch.randint(0, 512, (4, 2, 16)) for _ in range(2)]
    for step_ids in all_ids:
        ids = step_ids[dp_rank]
        loss = engine(ids, labels=ids)
        engine.backward(loss)
        engine.step()
        for d in range(4):
            l2 = ref(step_ids[d], labels=step_ids[d])
            (l2 / 4).backward()
        opt_ref.step()
        opt_ref.zero_grad()

    for (n, p), (_, pr) in zip(engine.module.named_parameters(),
                               ref.named_parameters()):
        if getattr(p, "tensor_model_parallel", False):
            continue
        torch.testing.assert_close(p, pr, rtol=3e-4, atol=5e-4), n


def test_tp2_dp4_ws8():
    run_distributed(_tp2dp4_worker, world_size=8, timeout=600)


class _ManyParamNet(nn.Module):
    """~40 variable-size params to stress the ZeRO-3 prefetch window,
    inflight backpressure and release discipline at tight budgets."""

    def __init__(self, seed=5):
        super().__init__()
        torch.manual_seed(seed)
        sizes = [8, 64, 16, 128, 32, 8, 96, 64, 16, 48] * 2
        dims = []
        prev = 32
        for s in sizes:
            dims.append((prev, s))
            prev = s
        self.layers = nn.ModuleList(
            [nn.Linear(i, o, bias=(n % 3 != 0))
             for n, (i, o) in enumerate(dims)])
        self.out = nn.Linear(prev, 1)

    def forward(self, x, labels=None):
        for l in self.layers:
            x = torch.tanh(l(x))
        y = self.out(x)
        if labels is not None:
            return nn.functional.mse_loss(y.float(), labels.float())
        return y


def _z3_stress_worker(rank, world):
    import deepspeed_amd

    model = _ManyParamNet()
    ref = copy.deepcopy(model)
    engine, _, _, _ = deepspeed_amd.initialize(model=model, config={
        "train_micro_batch_size_per_gpu": 4,
        "zero_optimization": {
            "stage": 3,
            "overlap_comm": True,
            "stage3_prefetch_bucket_size": 2_000,       # tiny window
            "stage3_param_persistence_threshold": 10,   # nothing persists
            "stage3_max_live_parameters": 20_000,       # tight live budget
        },
        "optimizer": {"type": "AdamW", "params": {"lr": 1e-2}}})
    opt_ref = torch.optim.AdamW(ref.parameters(), lr=1e-2)
    torch.manual_seed(3)
    for _ in range(4):
        x = torch.randn(4, 32)
        y = torch.randn(4, 1)
        loss = engine(x, labels=y)
        engine.backward(loss)
        engine.step()
        l2 = ref(x, labels=y)
        l2.backward()
        opt_ref.step()
        opt_ref.zero_grad()
        assert abs(loss.item() - l2.item()) < 5e-3, (loss.item(), l2.item())
    sd = engine.optimizer.get_full_state_dict()
    if rank == 0:
        ref_sd = ref.state_dict()
        for name, t in sd.items():
            torch.testing.assert_close(t.float().cpu(), ref_sd[name].float(),
                                       rtol=2e-3, atol=2e-3)


def test_zero3_backpressure_stress_ws8():
    run_distributed(_z3_stress_worker, world_size=8, timeout=600)


def _pp2tp2dp2_worker(rank, world, zero_stage=0):
    """Full 3D: PP=2 x TP=2 x DP=2 on 8 ranks. Each pipeline stage holds
    TP-sharded blocks (column -> row parallel pair); parity against the
    unsharded sequential model with dp-averaged gradients."""
    import deepspeed_amd
    from deepspeed_amd.parallel import groups as pgroups
    from deepspeed_amd.runtime.pipe.module import PipelineModule, LayerSpec
    from deepspeed_amd.runtime.tensor_parallel.layers import (
        ColumnParallelLinear, RowParallelLinear)

    D, FF = 8, 16

    class RefBlock(nn.Module):
        def __init__(self, seed):
            super().__init__()
            torch.manual_seed(seed)
            self.fc1 = nn.Linear(D, FF, bias=False)
            self.fc2 = nn.Linear(FF, D, bias=False)

        def forward(self, x):
            return self.fc2(torch.relu(self.fc1(x)))

    class TPBlock(nn.Module):
        """Same math, fc1 column-sharded / fc2 row-sharded over TP."""

        def __init__(self, seed):
            super().__init__()
            ref = RefBlock(seed)
            g = pgroups.get_tensor_parallel_group()
            tp = pgroups.get_tensor_parallel_world_size()
            tr = pgroups.get_tensor_parallel_rank()
            w1 = ref.fc1.weight.detach().chunk(tp, dim=0)[tr].clone()
            w2 = ref.fc2.weight.detach().chunk(tp, dim=1)[tr].clone()
            self.col = ColumnParallelLinear(w1, None, g)
            self.row = RowParallelLinear(w2, None, g)

        def forward(self, x):
            return self.row(torch.relu(self.col(x)))

    cfg = {"train_micro_batch_size_per_gpu": 4,
           "gradient_accumulation_steps": 2,
           "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}}}
    if zero_stage:
        cfg["zero_optimization"] = {"stage": zero_stage,
                                    "overlap_comm": False}
    specs = [LayerSpec(TPBlock, 21), LayerSpec(TPBlock, 22),
             LayerSpec(TPBlock, 23), LayerSpec(TPBlock, 24)]
    net = PipelineModule(specs, num_stages=2, tp_size=2,
                         loss_fn=nn.functional.mse_loss,
                         partition_method="uniform")
    assert net.grid.data_parallel_size == 2
    assert net.grid.tensor_parallel_size == 2
    engine, _, _, _ = deepspeed_amd.initialize(model=net, config=cfg)

    def make_data(n, bs, seed):
        g = torch.Generator().manual_seed(seed)
        return [(torch.randn(bs, D, generator=g),
                 torch.randn(bs, D, generator=g)) for _ in range(n)]

    n_steps, mbs = 2, 2
    dp = net.grid.data_parallel_id
    data_all = [make_data(n_steps * mbs, 4, seed=77 + d) for d in range(2)]
    it = iter(data_all[dp])
    losses = [engine.train_batch(it).item() for _ in range(n_steps)]

    ref = nn.Sequential(*[RefBlock(s) for s in (21, 22, 23, 24)])
    opt = torch.optim.AdamW(ref.parameters(), lr=1e-3)
    its = [iter(data_all[d]) for d in range(2)]
    ref_losses = []
    for _ in range(n_steps):
        tot = 0.0
        for d in range(2):
            for _ in range(mbs):
                x, y = next(its[d])
                loss = nn.functional.mse_loss(ref(x), y)
                (loss / (mbs * 2)).backward()
                tot += loss.item()
        opt.step()
        opt.zero_grad()
        ref_losses.append(tot / (mbs * 2))
    for got, want in zip(losses, ref_losses):
        assert abs(got - want) < 1e-5, (losses, ref_losses)

    # my TP shards must match the reference slices after optimization
    tr = net.grid.tensor_parallel_id
    ref_blocks = list(ref)[net.part_start:net.part_end]
    for m, r in zip(net.forward_funcs, ref_blocks):
        torch.testing.assert_close(
            m.col.weight, r.fc1.weight.detach().chunk(2, dim=0)[tr],
            rtol=1e-4, atol=2e-5)
        torch.testing.assert_close(
            m.row.weight, r.fc2.weight.detach().chunk(2, dim=1)[tr],
            rtol=1e-4, atol=2e-5)


def test_pipeline_3d_pp2tp2dp2_ws8():
    run_distributed(_pp2tp2dp2_worker, world_size=8, timeout=600)


def test_pipeline_3d_zero1_ws8():
    """Same 3D mesh with ZeRO-1 partitioning over each (stage, tp) cell's
    DP replicas — exercises the combined pipe x tensor norm group."""
    run_distributed(_pp2tp2dp2_worker, world_size=8, timeout=600, args=(1,))


def _pp2tp2dp2_ckpt_worker(rank, world, tmp):
    """3D (PP2xTP2xDP2) checkpoint save/resume: model files are
    (mp_rank, pp_rank)-qualified so no two of the 4 model-owning groups
    collide; resume restores exact weights."""
    import deepspeed_amd
    from deepspeed_amd.parallel import groups as pgroups
    from deepspeed_amd.runtime.pipe.module import PipelineModule, LayerSpec
    from deepspeed_amd.runtime.tensor_parallel.layers import (
        ColumnParallelLinear, RowParallelLinear)

    D, FF = 8, 16

    class TPBlock(nn.Module):
        def __init__(self, seed):
            super().__init__()
            torch.manual_seed(seed)
            w1 = torch.randn(FF, D)
            w2 = torch.randn(D, FF)
            g = pgroups.get_tensor_parallel_group()
            tp = pgroups.get_tensor_parallel_world_size()
            tr = pgroups.get_tensor_parallel_rank()
            self.col = ColumnParallelLinear(
                w1.chunk(tp, dim=0)[tr].clone(), None, g)
            self.row = RowParallelLinear(
                w2.chunk(tp, dim=1)[tr].clone(), None, g)

        def forward(self, x):
            return self.row(torch.relu(self.col(x)))

    def build():
        net = PipelineModule([LayerSpec(TPBlock, 31), LayerSpec(TPBlock, 32),
                              LayerSpec(TPBlock, 33), LayerSpec(TPBlock, 34)],
                             num_stages=2, tp_size=2,
                             loss_fn=nn.functional.mse_loss,
                             partition_method="uniform")
        eng, _, _, _ = deepspeed_amd.initialize(model=net, config={
            "train_micro_batch_size_per_gpu": 4,
            "gradient_accumulation_steps": 2,
            "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}}})
        return net, eng

    def data(n, seed):
        g = torch.Generator().manual_seed(seed)
        return [(torch.randn(4, D, generator=g),
                 torch.randn(4, D, generator=g)) for _ in range(n)]

    net, eng = build()
    dp = net.grid.data_parallel_id
    it = iter(data(6, seed=99 + dp))
    for _ in range(2):
        eng.train_batch(it)
    eng.save_checkpoint(tmp, tag="t3d")
    cont = eng.train_batch(it).item()

    net2, eng2 = build()
    tag, _ = eng2.load_checkpoint(tmp, tag="t3d")
    assert tag is not None
    it2 = iter(data(6, seed=99 + dp))
    for _ in range(4):
        next(it2)
    resumed = eng2.train_batch(it2).item()
    assert abs(resumed - cont) < 1e-6, (resumed, cont)


def test_pipeline_3d_checkpoint_resume_ws8(tmp_path):
    run_distributed(_pp2tp2dp2_ckpt_worker, world_size=8, timeout=600,
                    args=(str(tmp_path),))
