"""Aux subsystem tests: elasticity math, curriculum scheduler, progressive
layer drop, eigenvalue power iteration, engine.compile
(reference contracts: tests/unit/elasticity/test_elastic.py,
tests/unit/runtime/test_data_efficiency.py, tests/unit/runtime/test_pld.py).
"""

import math

import pytest
import torch


def test_elastic_config_basic():
    from deepspeed_amd.elasticity import compute_elastic_config
    ds_config = {"elasticity": {
        "enabled": True, "max_train_batch_size": 2000,
        "micro_batch_sizes": [2, 4, 6], "min_gpus": 1, "max_gpus": 10000,
        "prefer_larger_batch": True}}
    batch, gpus = compute_elastic_config(ds_config)
    assert batch <= 2000 and batch % 2 == 0
    # every valid gpu count divides batch by some micro batch
    for g in gpus[:20]:
        assert any(batch % (m * g) == 0 for m in (2, 4, 6))


def test_elastic_config_world_size():
    from deepspeed_amd.elasticity import (ElasticityError,
                                          compute_elastic_config)
    ds_config = {"elasticity": {
        "enabled": True, "max_train_batch_size": 100,
        "micro_batch_sizes": [4], "min_gpus": 1, "max_gpus": 16}}
    batch, gpus, micro = compute_elastic_config(ds_config, world_size=4,
                                                return_microbatch=True)
    assert 4 in gpus and batch % (micro * 4) == 0
    with pytest.raises(ElasticityError):
        compute_elastic_config({"elasticity": {"enabled": False}})


def test_curriculum_fixed_linear():
    from deepspeed_amd.runtime.data_pipeline import CurriculumScheduler
    s = CurriculumScheduler({
        "curriculum_type": "fixed_linear", "min_difficulty": 8,
        "max_difficulty": 64,
        "schedule_config": {"total_curriculum_step": 100,
                            "difficulty_step": 8}})
    assert s.update_difficulty(0) == 8
    mid = s.update_difficulty(50)
    assert 8 < mid < 64 and mid % 8 == 0
    assert s.update_difficulty(100) == 64
    assert s.update_difficulty(10_000) == 64


def test_curriculum_fixed_discrete():
    from deepspeed_amd.runtime.data_pipeline import CurriculumScheduler
    s = CurriculumScheduler({
        "curriculum_type": "fixed_discrete", "min_difficulty": 2,
        "max_difficulty": 10,
        "schedule_config": {"difficulty": [2, 6, 10],
                            "max_step": [10, 20]}})
    assert s.update_difficulty(5) == 2
    assert s.update_difficulty(15) == 6
    assert s.update_difficulty(25) == 10


def test_progressive_layer_drop():
    from deepspeed_amd.runtime.progressive_layer_drop import \
        ProgressiveLayerDrop
    pld = ProgressiveLayerDrop(theta=0.5, gamma=0.001)
    assert pld.get_theta() == 1.0
    pld.update_state(0)
    assert abs(pld.get_theta() - 1.0) < 1e-6
    pld.update_state(10_000)
    assert 0.5 <= pld.get_theta() < 0.51  # decays toward theta
    assert pld.get_state()["progressive_layer_drop"]


def test_eigenvalue_power_iteration():
    from deepspeed_amd.runtime.eigenvalue import Eigenvalue
    torch.manual_seed(0)
    lin = torch.nn.Linear(8, 1, bias=False)
    lin._deepspeed_eigenvalue_block = True
    X = torch.randn(256, 8)
    loss = (lin(X) ** 2).mean()
    loss.backward(create_graph=True)
    ev = Eigenvalue(max_iter=500, tol=1e-5).compute_eigenvalue(lin)
    # quadratic loss: Hessian = 2/N X^T X; top eigenvalue known
    H = 2 * X.T @ X / X.shape[0]
    expect = torch.linalg.eigvalsh(H).max().item()
    assert len(ev) == 1
    assert abs(ev[0] - expect) / expect < 0.05


def test_engine_compile_flag():
    from .common import run_local

    def worker(rank, world):
        import deepspeed_amd
        from deepspeed_amd.models import GPT2ForCausalLM, gpt2_tiny
        model = GPT2ForCausalLM(gpt2_tiny())
        engine, _, _, _ = deepspeed_amd.initialize(model=model, config={
            "train_micro_batch_size_per_gpu": 1,
            "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}}})
        assert not engine.is_compiled
        engine.compile(backend="eager")  # inductor needs a compiler rig
        assert engine.is_compiled
        ids = torch.randint(0, 128, (1, 16))
        loss = engine(ids, labels=ids)
        engine.backward(loss)
        engine.step()
        assert torch.isfinite(loss)

    run_local(worker)


def test_curriculum_data_sampler():
    from deepspeed_amd.runtime.data_pipeline.curriculum_scheduler import \
        CurriculumScheduler
    from deepspeed_amd.runtime.data_pipeline.data_sampler import \
        DeepSpeedDataSampler
    diffs = [10, 20, 30, 40] * 25  # 100 samples
    sched = CurriculumScheduler({
        "curriculum_type": "fixed_discrete", "min_difficulty": 10,
        "max_difficulty": 40,
        "schedule_config": {"difficulty": [10, 40], "max_step": [1]}})
    s = DeepSpeedDataSampler(diffs, sched, batch_size=4, dp_rank=0, dp_size=2)
    batches = list(s)
    assert all(len(b) == 4 for b in batches)
    # first batches: only difficulty-10 samples are eligible
    assert all(diffs[i] == 10 for i in batches[0])
    # later batches include harder samples (difficulty opens at step 2)
    assert any(diffs[i] > 10 for b in batches[2:] for i in b)
    # DP disjointness: rank 1 gets different indices for the same epoch/step
    s1 = DeepSpeedDataSampler(diffs, CurriculumScheduler({
        "curriculum_type": "fixed_discrete", "min_difficulty": 10,
        "max_difficulty": 40,
        "schedule_config": {"difficulty": [10, 40], "max_step": [1]}}),
        batch_size=4, dp_rank=1, dp_size=2)
    b1 = next(iter(s1))
    assert not set(batches[0]) & set(b1)


def test_checkpoint_engines(tmp_path):
    from deepspeed_amd.runtime.checkpoint_engine import (
        AsyncTorchCheckpointEngine, TorchCheckpointEngine,
        create_checkpoint_engine)
    sd = {"w": torch.randn(100), "step": 7}
    for eng in (TorchCheckpointEngine(), AsyncTorchCheckpointEngine()):
        p = str(tmp_path / f"{type(eng).__name__}.pt")
        eng.create("tag")
        eng.save(sd, p)
        assert eng.commit("tag")
        back = eng.load(p)
        assert torch.equal(back["w"], sd["w"]) and back["step"] == 7
    assert isinstance(create_checkpoint_engine("async"),
                      AsyncTorchCheckpointEngine)


def test_random_ltd():
    from deepspeed_amd.runtime.data_pipeline.random_ltd import (
        RandomLayerTokenDrop, RandomLTDScheduler, convert_to_random_ltd)
    import torch.nn as nn

    sched = RandomLTDScheduler(total_layers=4, random_ltd_layer_num=2,
                               start_seq=8, max_seq=16, step_size=4,
                               schedule_steps=10)
    assert sched.update_seq(0) == 8
    assert sched.update_seq(5) == 12
    assert sched.update_seq(10) == 16
    sched.current_seq = 8

    class Layers(nn.Module):
        def __init__(self):
            super().__init__()
            self.layers = nn.ModuleList([nn.Linear(4, 4) for _ in range(4)])

    m = Layers()
    n = convert_to_random_ltd(m, "layers", sched)
    assert n == 2 and isinstance(m.layers[1], RandomLayerTokenDrop)

    x = torch.randn(2, 16, 4)
    m.train()
    y = m.layers[1](x)
    assert y.shape == x.shape
    # dropped positions pass through unchanged: exactly seq-keep untouched
    untouched = (y == x).all(dim=-1).sum(dim=1)
    assert (untouched == 16 - 8).all()
    # eval mode: full sequence processed
    m.eval()
    y2 = m.layers[1](x)
    assert not torch.equal(y2, x)


def test_data_analyzer_end_to_end(tmp_path):
    from deepspeed_amd.runtime.data_pipeline.data_analyzer import (
        DataAnalyzer, load_index_to_metric, metric_seqlen)
    from deepspeed_amd.runtime.data_pipeline import CurriculumScheduler
    from deepspeed_amd.runtime.data_pipeline.data_sampler import \
        DeepSpeedDataSampler

    torch.manual_seed(0)
    data = [torch.randint(0, 100, (int(l),))
            for l in torch.randint(4, 33, (40,))]
    # 2-worker map, then reduce
    for w in range(2):
        DataAnalyzer(data, save_path=str(tmp_path), worker_id=w,
                     num_workers=2).run_map()
    merged = DataAnalyzer(data, save_path=str(tmp_path),
                          num_workers=2).run_reduce()
    diffs = merged["seqlen"]
    assert diffs == [float(len(s)) for s in data]
    assert load_index_to_metric(str(tmp_path), "seqlen") == diffs

    # feeds the curriculum sampler directly
    sched = CurriculumScheduler({
        "curriculum_type": "fixed_linear", "min_difficulty": 8,
        "max_difficulty": 32,
        "schedule_config": {"total_curriculum_step": 4,
                            "difficulty_step": 8}})
    sampler = DeepSpeedDataSampler(diffs, sched, batch_size=2)
    first = next(iter(sampler))
    assert all(diffs[i] <= 8 for i in first)


def test_sparse_attention_layouts():
    from deepspeed_amd.ops.sparse_attention import (BigBirdSparsityConfig,
                                                    DenseSparsityConfig,
                                                    FixedSparsityConfig,
                                                    SparseSelfAttention)
    import torch.nn.functional as F
    H, B, S, D, blk = 2, 2, 64, 8, 16
    q = torch.randn(B, H, S, D)
    k = torch.randn(B, H, S, D)
    v = torch.randn(B, H, S, D)

    # dense config == plain attention
    dense = SparseSelfAttention(DenseSparsityConfig(H, blk))
    torch.testing.assert_close(dense(q, k, v),
                               F.scaled_dot_product_attention(q, k, v))

    fixed = FixedSparsityConfig(H, blk, num_local_blocks=2,
                                attention="unidirectional")
    layout = fixed.make_layout(S)
    n = S // blk
    assert layout.shape == (H, n, n)
    assert not layout[:, 0, 1].any()          # causal: no future blocks
    assert layout[:, 3, 2].all()              # local window
    out = SparseSelfAttention(fixed)(q, k, v)
    assert out.shape == q.shape and torch.isfinite(out).all()
    # causal masked-out blocks change nothing: perturb a far-future key
    k2 = k.clone()
    k2[:, :, -1] += 100.0
    out2 = SparseSelfAttention(fixed)(q, k2, v)
    torch.testing.assert_close(out[:, :, :16], out2[:, :, :16])

    bb = BigBirdSparsityConfig(H, blk, num_random_blocks=1)
    lb = bb.make_layout(S)
    assert lb[:, 2, 1].all() and lb[:, 2, 3].all()   # sliding window
    assert lb[:, :, 0].all() and lb[:, 0, :].all()   # global block


def test_zero_memory_estimators():
    from deepspeed_amd.runtime.zero import (
        estimate_zero2_model_states_mem_needs,
        estimate_zero3_model_states_mem_needs)
    P = 8_000_000_000
    gpu3, cpu3 = estimate_zero3_model_states_mem_needs(P, 8, 1)
    # 8B over 8 GPUs: (2+2+12)*1e9*1.5 = 24 GB/GPU
    assert abs(gpu3 - 24e9) / 24e9 < 0.01 and cpu3 == 0
    gpu3o, cpu3o = estimate_zero3_model_states_mem_needs(P, 8, 1,
                                                         cpu_offload=True)
    assert gpu3o < gpu3 and cpu3o > 100e9  # states land in host DRAM
    gpu2, _ = estimate_zero2_model_states_mem_needs(P, 8, 1)
    assert gpu2 > gpu3  # full replicas of params+grads


def test_deepspeed_transformer_layer():
    from deepspeed_amd import (DeepSpeedTransformerConfig,
                               DeepSpeedTransformerLayer)
    torch.manual_seed(0)
    cfg = DeepSpeedTransformerConfig(hidden_size=64, intermediate_size=128,
                                     heads=4, attn_dropout_ratio=0.0,
                                     hidden_dropout_ratio=0.0,
                                     pre_layer_norm=True, seed=11)
    layer = DeepSpeedTransformerLayer(cfg)
    x = torch.randn(2, 16, 64, requires_grad=True)
    mask = torch.ones(2, 16, dtype=torch.long)
    y = layer(x, mask)
    assert y.shape == x.shape
    y.sum().backward()
    assert x.grad is not None and torch.isfinite(x.grad).all()
    # post-LN variant runs too
    cfg2 = DeepSpeedTransformerConfig(hidden_size=64, intermediate_size=128,
                                      heads=4, pre_layer_norm=False,
                                      attn_dropout_ratio=0.0,
                                      hidden_dropout_ratio=0.0)
    y2 = DeepSpeedTransformerLayer(cfg2)(x)
    assert y2.shape == x.shape


def test_flash_bwd_reference_matches_autograd():
    """The tile-level flash-backward algebra (the HIP kernels' blueprint)
    must reproduce autograd's SDPA gradients exactly (fp32, causal, GQA,
    ragged tail)."""
    import torch.nn.functional as F
    from deepspeed_amd.ops.flash_bwd_ref import flash_bwd_reference

    torch.manual_seed(0)
    B, H, Hkv, S, D = 2, 4, 2, 70, 16  # S=70: ragged last tile
    q = torch.randn(B, H, S, D, requires_grad=True)
    k = torch.randn(B, Hkv, S, D, requires_grad=True)
    v = torch.randn(B, Hkv, S, D, requires_grad=True)
    kr = k.repeat_interleave(2, 1)
    vr = v.repeat_interleave(2, 1)
    o = F.scaled_dot_product_attention(q, kr, vr, is_causal=True)
    do = torch.randn_like(o)
    o.backward(do)

    # lse computed exactly as the fwd kernel defines it
    scale = 1.0 / (D ** 0.5)
    st = (q.detach().float() @ kr.detach().float().transpose(-1, -2)) * scale
    mask = torch.triu(torch.ones(S, S, dtype=torch.bool), 1)
    st = st.masked_fill(mask, -float("inf"))
    lse = torch.logsumexp(st, dim=-1)

    dq, dk, dv = flash_bwd_reference(q.detach(), k.detach(), v.detach(),
                                     o.detach(), do, lse, causal=True,
                                     tile=32)
    torch.testing.assert_close(dq, q.grad, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(dk, k.grad, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(dv, v.grad, rtol=1e-4, atol=1e-5)


def test_flash_bwd_lds_transpose_maps():
    """Lane-exact simulation of attention_bwd.hip's C/D->A LDS transposes
    (the only index machinery not shared with the GPU-validated forward)."""
    import numpy as np

    def cd_row(r, half):
        return (r & 3) + 8 * (r >> 2) + 4 * half

    T = 32
    PT = np.arange(T * T).reshape(T, T)  # P^T[kv][q] distinct values

    # C/D fragments: lane holds col=lane&31, rows cd_row(r, lane>>5)
    lds = np.zeros((T, T), dtype=int)
    for lane in range(64):
        col, half = lane & 31, lane >> 5
        for r in range(16):
            lds[cd_row(r, half), col] = PT[cd_row(r, half), col]
    assert (lds == PT).all()

    # kernel-1 A-frag read: row=lane&31 (kv), k-slot q = kk*16+8*half+[0,8)
    for lane in range(64):
        row, half = lane & 31, lane >> 5
        for kk in range(2):
            frag = lds[row, kk * 16 + 8 * half: kk * 16 + 8 * half + 8]
            assert (frag == PT[row, kk * 16 + 8 * half:
                               kk * 16 + 8 * half + 8]).all()

    # kernel-2 transposed store: lds2[q][kv] then read row=q, k=kv
    lds2 = np.zeros((T, T), dtype=int)
    for lane in range(64):
        col, half = lane & 31, lane >> 5
        for r in range(16):
            lds2[col, cd_row(r, half)] = PT[cd_row(r, half), col]
    assert (lds2 == PT.T).all()
    for lane in range(64):
        row, half = lane & 31, lane >> 5
        for kk in range(2):
            frag = lds2[row, kk * 16 + 8 * half: kk * 16 + 8 * half + 8]
            assert (frag == PT.T[row, kk * 16 + 8 * half:
                                 kk * 16 + 8 * half + 8]).all()


def test_flash_bwd_address_math():
    """Mirror attention_bwd.hip's pointer arithmetic in Python and check
    every fragment load/store hits the intended (b,h,s,d) element."""
    B, H, Hkv, S, D = 2, 4, 2, 64, 128
    G = H // Hkv

    def cd_row(r, half):
        return (r & 3) + 8 * (r >> 2) + 4 * half

    def bhsd(b, h, s, d, HH):  # flat index of [B, HH, S, D]
        return ((b * HH + h) * S + s) * D + d

    def bhds(b, h, d, s, HH):  # flat index of [B, HH, D, S] (qt/kt/dot)
        return ((b * HH + h) * D + d) * S + s

    b, hkv, kv0 = 1, 1, 32
    for lane in range(0, 64, 17):
        col, half = lane & 31, lane >> 5
        kvrow = kv0 + col
        # K/V A-frag base (kernel 1)
        base = ((b * Hkv + hkv) * S + kvrow) * D + 8 * half
        for kk in range(8):
            for reg in range(8):
                want = bhsd(b, hkv, kvrow, kk * 16 + 8 * half + reg, Hkv)
                assert base + kk * 16 + reg == want
        for g in range(G):
            h = hkv * G + g
            qbase = ((b * H + h) * S) * D
            tbase = ((b * H + h) * D) * S
            sbase = (b * H + h) * S
            qs = 32
            qrow = qs + col
            # Q/dO B-frags
            for kk in range(8):
                for reg in range(8):
                    assert qbase + qrow * D + kk * 16 + 8 * half + reg == \
                        bhsd(b, h, qrow, kk * 16 + 8 * half + reg, H)
            # lse/delta scalars
            assert sbase + qrow == (b * H + h) * S + qrow
            # dot/qt B-frags for dV/dK: (col=d, k=q)
            for dblk in range(4):
                trow = (dblk * 32 + col) * S + qs
                for kk in range(2):
                    for reg in range(8):
                        q_idx = qs + kk * 16 + 8 * half + reg
                        assert tbase + trow + kk * 16 + 8 * half + reg == \
                            bhds(b, h, dblk * 32 + col, q_idx, H)
        # dk/dv writes: (col=d_local, row=kv)
        obase = ((b * Hkv + hkv) * S) * D
        for dblk in range(4):
            for r in range(16):
                kvl = cd_row(r, half)
                assert obase + (kv0 + kvl) * D + dblk * 32 + col == \
                    bhsd(b, hkv, kv0 + kvl, dblk * 32 + col, Hkv)

    # kernel 2: kt B-frag (col=d, k=kv) and dq writes
    h, qs = 3, 32
    hkv2 = h // G
    ktbase = ((b * Hkv + hkv2) * D) * S
    for lane in (0, 33, 63):
        col, half = lane & 31, lane >> 5
        for dblk in range(4):
            kv0b = 0
            trow = ktbase + (dblk * 32 + col) * S + kv0b
            for kk in range(2):
                for reg in range(8):
                    kv_idx = kv0b + kk * 16 + 8 * half + reg
                    assert trow + kk * 16 + 8 * half + reg == \
                        bhds(b, hkv2, dblk * 32 + col, kv_idx, Hkv)
        qbase = ((b * H + h) * S) * D
        for dblk in range(4):
            for r in range(16):
                ql = cd_row(r, half)
                assert qbase + (qs + ql) * D + dblk * 32 + col == \
                    bhsd(b, h, qs + ql, dblk * 32 + col, H)


def test_fused_ce_fallback_matches_fp32_path():
    """CPU fallback of fused_cross_entropy_sum == the original fp32 chunked
    CE math (this path feeds every existing engine test)."""
    from deepspeed_amd.ops.cross_entropy import fused_cross_entropy_sum
    torch.manual_seed(0)
    logits = torch.randn(64, 100)
    labels = torch.randint(0, 100, (64,))
    labels[::5] = -100
    l, c = fused_cross_entropy_sum(logits, labels)
    ref = torch.nn.functional.cross_entropy(logits.float(), labels,
                                            ignore_index=-100,
                                            reduction="sum")
    torch.testing.assert_close(l, ref)
    assert c.item() == (labels != -100).sum().item()


def test_flops_profiler_counts_gemms():
    from deepspeed_amd.profiling.flops_profiler import FlopsProfiler
    import torch.nn as nn
    m = nn.Sequential(nn.Linear(64, 128), nn.ReLU(), nn.Linear(128, 10))
    prof = FlopsProfiler(m)
    prof.start_profile()
    x = torch.randn(4, 64)
    m(x)
    flops = prof.get_total_flops()
    params = prof.get_total_params()
    prof.end_profile()
    # 2*B*(64*128 + 128*10) MACs->flops
    expect = 2 * 4 * (64 * 128 + 128 * 10)
    assert abs(flops - expect) / expect < 0.05, (flops, expect)
    assert params == 64 * 128 + 128 + 128 * 10 + 10


def test_evoformer_attention_reference_math():
    """DS4Sci_EvoformerAttention vs an index-loop fp32 reference with both
    bias patterns (mask bias [B,N,1,1,S] + pair bias [B,1,H,S,S])."""
    import math
    from deepspeed_amd.ops.evoformer import DS4Sci_EvoformerAttention
    torch.manual_seed(0)
    B, N, S, H, D = 2, 3, 5, 2, 4
    Q, K, V = (torch.randn(B, N, S, H, D) for _ in range(3))
    bias1 = torch.randn(B, N, 1, 1, S) * 0.5
    bias2 = torch.randn(B, 1, H, S, S) * 0.5
    out = DS4Sci_EvoformerAttention(Q, K, V, [bias1, bias2])
    assert out.shape == (B, N, S, H, D)
    scale = 1.0 / math.sqrt(D)
    for b in range(B):
        for n in range(N):
            for h in range(H):
                for s in range(S):
                    logit = (Q[b, n, s, h] @ K[b, n, :, h].T) * scale \
                        + bias1[b, n, 0, 0] + bias2[b, 0, h, s]
                    p = torch.softmax(logit, -1)
                    ref = p @ V[b, n, :, h]
                    torch.testing.assert_close(out[b, n, s, h], ref,
                                               atol=1e-5, rtol=1e-5)
    # no-bias and single-bias paths
    out_nb = DS4Sci_EvoformerAttention(Q, K, V)
    out_b1 = DS4Sci_EvoformerAttention(Q, K, V, [bias1, None])
    assert not torch.allclose(out_nb, out_b1)


def _safe_mode_worker(rank, world):
    from deepspeed_amd.utils.safe_mode import (
        assert_ints_same_as_other_ranks, enable_safe_mode, checked)
    # identical sequences pass
    assert_ints_same_as_other_ranks([1, 2, 3], what="ok-case")
    # divergent sequences raise on every rank, naming the position
    import pytest
    with pytest.raises(RuntimeError, match="diverges|length"):
        assert_ints_same_as_other_ranks([1, 2, 3 + rank], what="bad-case")
    # checked() is a no-op until enabled
    checked([rank], what="gated")
    enable_safe_mode(True)
    try:
        with pytest.raises(RuntimeError):
            checked([rank], what="gated-on")
    finally:
        enable_safe_mode(False)


def test_safe_mode_cross_rank_asserts():
    from .common import run_distributed
    run_distributed(_safe_mode_worker, world_size=2)


def test_safe_mode_zero3_trace():
    """ZeRO-3 end-to-end with safe mode on: consistent ranks pass."""
    import os

    def worker(rank, world):
        os.environ["DS_AMD_SAFE_MODE"] = "1"
        import importlib
        from deepspeed_amd.utils import safe_mode
        importlib.reload(safe_mode)
        import deepspeed_amd
        import torch
        from deepspeed_amd.models import LlamaForCausalLM, llama_tiny
        cfg = llama_tiny()
        torch.manual_seed(4)
        model = LlamaForCausalLM(cfg)
        engine, _, _, _ = deepspeed_amd.initialize(model=model, config={
            "train_micro_batch_size_per_gpu": 2,
            "zero_optimization": {"stage": 3},
            "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}}})
        for _ in range(2):
            ids = torch.randint(0, cfg.vocab_size, (2, 16))
            loss = engine(ids, labels=ids)
            engine.backward(loss)
            engine.step()

    from .common import run_distributed
    run_distributed(worker, world_size=2)


def test_data_analyzer_parallel_map_reduce(tmp_path):
    from deepspeed_amd.runtime.data_pipeline.data_analyzer import (
        run_analysis_parallel, load_index_to_metric)
    data = [list(range(3 + i % 7)) for i in range(40)]
    out = run_analysis_parallel(data, save_path=str(tmp_path), num_workers=3)
    assert out["seqlen"] == [len(s) for s in data]
    assert load_index_to_metric(str(tmp_path), "seqlen") == out["seqlen"]


def test_curriculum_dataloader_from_config():
    """data_efficiency.curriculum_learning drives the engine dataloader:
    early batches contain only short samples, longer appear as the
    difficulty schedule opens up."""
    from .common import run_local

    def worker(rank, world):
        import torch
        import deepspeed_amd

        class ToyDS(torch.utils.data.Dataset):
            def __len__(self):
                return 64

            def __getitem__(self, i):
                n = 4 + (i % 8)
                return torch.full((n,), i, dtype=torch.long), n

        model = torch.nn.Linear(4, 4)
        engine, _, _, _ = deepspeed_amd.initialize(model=model, config={
            "train_micro_batch_size_per_gpu": 4,
            "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}},
            "data_efficiency": {
                "enabled": True,
                "curriculum_learning": {
                    "enabled": True, "curriculum_type": "fixed_linear",
                    "min_difficulty": 5, "max_difficulty": 12,
                    "schedule_config": {"total_curriculum_step": 8,
                                        "difficulty_step": 1}}}})
        dl = engine.deepspeed_io(ToyDS(), collate_fn=lambda b: b)
        lens_per_batch = []
        for batch in dl:
            lens_per_batch.append(max(n for _, n in batch))
            if len(lens_per_batch) >= 8:
                break
        assert lens_per_batch[0] <= 5, lens_per_batch
        assert max(lens_per_batch) > 5, lens_per_batch
        assert engine.curriculum_scheduler is not None

    run_local(worker)


def test_contiguous_allocator_defrag():
    import torch
    from deepspeed_amd.runtime.zero.contiguous_allocator import (
        ContiguousAllocator)
    a = ContiguousAllocator(10240, dtype=torch.float32, device="cpu")
    hs = [a.allocate(1500) for _ in range(6)]
    for i, h in enumerate(hs):
        h.tensor.fill_(float(i))
    # free alternating -> fragmented: 3 live x 1500, but gaps of 1536
    for h in hs[::2]:
        h.release()
    assert a.largest_free_block() < 4096
    # a 4000-elem allocation only fits after defragmentation
    big = a.allocate(4000)          # triggers defragment() internally
    big.tensor.fill_(9.0)
    # survivors kept their contents through the migration
    for i, h in zip((1, 3, 5), hs[1::2]):
        assert torch.all(h.tensor == float(i)), i
    assert torch.all(big.tensor == 9.0)
    # overlapping down-move correctness: one huge survivor shifted by less
    # than its own length
    b = ContiguousAllocator(8192, dtype=torch.float32, device="cpu")
    small = b.allocate(512)
    payload = b.allocate(6000)
    payload.tensor.copy_(torch.arange(6000, dtype=torch.float32))
    small.release()
    b.defragment()
    assert torch.equal(payload.tensor,
                       torch.arange(6000, dtype=torch.float32))


def test_fp16_auto_cast_inputs():
    """fp16.auto_cast: fp32 float inputs are cast to the engine dtype at
    the engine boundary (reference fp16 auto_cast)."""
    from .common import run_local
    run_local(_auto_cast_worker)


def _auto_cast_worker(rank=0, world=1):
    import deepspeed_amd

    seen = {}

    class Probe(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.fc = torch.nn.Linear(8, 1)

        def forward(self, x, labels=None):
            seen["dtype"] = x.dtype
            y = self.fc(x)
            return torch.nn.functional.mse_loss(y.float(),
                                                labels.float())

    eng, _, _, _ = deepspeed_amd.initialize(model=Probe(), config={
        "train_micro_batch_size_per_gpu": 2,
        "fp16": {"enabled": True, "auto_cast": True,
                 "initial_scale_power": 4},
        "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}}})
    loss = eng(torch.randn(2, 8), labels=torch.randn(2, 1))
    assert seen["dtype"] == torch.float16
    eng.backward(loss)
    eng.step()


def test_save_16bit_model_zero3_gate(tmp_path):
    """stage3_gather_16bit_weights_on_model_save=false refuses the
    consolidated save instead of silently gathering."""
    from .common import run_local
    run_local(_save16_gate_worker, args=(str(tmp_path),))


def _save16_gate_worker(rank, world, tmp_path):
    from pathlib import Path
    tmp_path = Path(tmp_path)
    import deepspeed_amd

    class M(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.fc = torch.nn.Linear(8, 1)

        def forward(self, x, labels=None):
            return torch.nn.functional.mse_loss(
                self.fc(x).float(), labels.float())

    def build(gather):
        return deepspeed_amd.initialize(model=M(), config={
            "train_micro_batch_size_per_gpu": 2,
            "bf16": {"enabled": True},
            "zero_optimization": {
                "stage": 3,
                "stage3_gather_16bit_weights_on_model_save": gather},
            "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}}})[0]

    eng = build(False)
    assert eng.save_16bit_model(str(tmp_path)) is False
    assert not (tmp_path / "pytorch_model.bin").exists()
    eng = build(True)
    assert eng.save_16bit_model(str(tmp_path)) is True
    sd = torch.load(tmp_path / "pytorch_model.bin", weights_only=True)
    assert "fc.weight" in sd and sd["fc.weight"].shape == (1, 8)


def test_random_ltd_from_config():
    """data_efficiency.random_ltd wraps middle decoder layers and the
    kept-token count follows the schedule across engine steps."""
    from .common import run_local
    run_local(_random_ltd_worker)


def _random_ltd_worker(rank=0, world=1):
    import deepspeed_amd
    from deepspeed_amd.models import LlamaForCausalLM, llama_tiny
    from deepspeed_amd.runtime.data_pipeline.random_ltd import \
        RandomLayerTokenDrop

    torch.manual_seed(0)
    model = LlamaForCausalLM(llama_tiny())
    engine, _, _, _ = deepspeed_amd.initialize(model=model, config={
        "train_micro_batch_size_per_gpu": 2,
        "bf16": {"enabled": True},
        "zero_optimization": {"stage": 2, "overlap_comm": False},
        "data_efficiency": {
            "enabled": True,
            "random_ltd": {"enabled": True, "layers_attr": "model.layers",
                           "skip_first": 1, "skip_last": 0,
                           "min_value": 8, "max_value": 32,
                           "seq_per_step": 8, "total_ltd_steps": 4}},
        "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}}})
    wrapped = [m for m in engine.module.modules()
               if isinstance(m, RandomLayerTokenDrop)]
    assert len(wrapped) == 1  # 2 layers, skip_first=1
    seqs = []
    for _ in range(5):
        ids = torch.randint(0, 500, (2, 32))
        loss = engine(ids.to(engine.device), labels=ids.to(engine.device))
        engine.backward(loss)
        engine.step()
        seqs.append(engine.random_ltd_scheduler.current_seq)
        assert torch.isfinite(loss)
    assert seqs[0] < seqs[-1] and seqs[-1] == 32, seqs


def test_progressive_layer_drop_from_config():
    from .common import run_local
    run_local(_pld_worker)


def _pld_worker(rank=0, world=1):
    import deepspeed_amd

    class M(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.fc = torch.nn.Linear(8, 1)

        def forward(self, x, labels=None):
            return torch.nn.functional.mse_loss(self.fc(x).float(),
                                                labels.float())

    eng, _, _, _ = deepspeed_amd.initialize(model=M(), config={
        "train_micro_batch_size_per_gpu": 2,
        "progressive_layer_drop": {"enabled": True, "theta": 0.5,
                                   "gamma": 0.1},
        "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}}})
    thetas = [eng.progressive_layer_drop.get_theta()]
    for _ in range(3):
        loss = eng(torch.randn(2, 8).to(eng.device),
                   labels=torch.randn(2, 1).to(eng.device))
        eng.backward(loss)
        eng.step()
        thetas.append(eng.progressive_layer_drop.get_theta())
    # theta decays from 1.0 toward theta=0.5
    assert thetas[0] == 1.0 and all(a > b for a, b in zip(thetas, thetas[1:]))
    assert thetas[-1] > 0.5


def test_engine_dynamic_batch_api():
    from .common import run_local
    run_local(_dyn_batch_worker)


def _dyn_batch_worker(rank=0, world=1):
    import deepspeed_amd

    class M(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.fc = torch.nn.Linear(8, 1)

        def forward(self, x, labels=None):
            return torch.nn.functional.mse_loss(self.fc(x).float(),
                                                labels.float())

    eng, _, _, _ = deepspeed_amd.initialize(model=M(), config={
        "train_micro_batch_size_per_gpu": 2,
        "gradient_accumulation_steps": 2,
        "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}}})
    assert eng.train_batch_size == 4
    eng.set_train_batch_size(8)  # gas 2 -> 4
    assert eng.gradient_accumulation_steps == 4
    import pytest as _pytest
    with _pytest.raises(ValueError):
        eng.set_train_batch_size(7)
    # train one full accumulation window under the new gas
    for i in range(4):
        loss = eng(torch.randn(2, 8).to(eng.device),
                   labels=torch.randn(2, 1).to(eng.device))
        eng.backward(loss)
        eng.step()
    assert eng.global_steps == 1
    assert eng.get_mom()[0] == (0.9, 0.999)
    assert eng.get_pld_theta() is None


def test_engine_config_accessors():
    from .common import run_local
    run_local(_accessor_worker)


def _accessor_worker(rank=0, world=1):
    import deepspeed_amd

    class M(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.fc = torch.nn.Linear(8, 1)

        def forward(self, x, labels=None):
            return torch.nn.functional.mse_loss(self.fc(x).float(),
                                                labels.float())

    eng, _, _, _ = deepspeed_amd.initialize(model=M(), config={
        "train_micro_batch_size_per_gpu": 2,
        "bf16": {"enabled": True},
        "gradient_clipping": 0.7,
        "zero_optimization": {"stage": 2, "overlap_comm": False,
                              "reduce_bucket_size": 1234},
        "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}}})
    assert eng.bfloat16_enabled() and not eng.fp16_enabled()
    assert eng.zero_optimization_stage() == 2
    assert eng.zero_optimization_partition_gradients()
    assert not eng.zero_optimization_partition_weights()
    assert eng.zero_reduce_bucket_size() == 1234
    assert eng.gradient_clipping() == 0.7
    assert eng.optimizer_name() == "AdamW"
    assert eng.get_batch_info() == (2, 2, 1)
    assert eng.was_step_applied()
    loss = eng(torch.randn(2, 8).to(eng.device).bfloat16(),
               labels=torch.randn(2, 1).to(eng.device))
    eng.backward(loss)
    eng.step()
    eng.empty_partition_cache()
    eng.zero_grad()
    eng.destroy()
    assert eng.optimizer is None


def test_curriculum_custom_schedule_and_post_process():
    from .common import run_local
    run_local(_curriculum_custom_worker)


def _curriculum_custom_worker(rank=0, world=1):
    import deepspeed_amd

    class DS(torch.utils.data.Dataset):
        def __len__(self):
            return 64

        def __getitem__(self, i):
            return torch.arange(4 + (i % 13)), i

    class M(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.fc = torch.nn.Linear(4, 1)

        def forward(self, x, labels=None):
            return self.fc(x.float()[:, :4]).sum()

    eng, _, _, _ = deepspeed_amd.initialize(model=M(), config={
        "train_micro_batch_size_per_gpu": 4,
        "data_efficiency": {
            "enabled": True,
            "curriculum_learning": {"enabled": True,
                                    "curriculum_type": "custom",
                                    "min_difficulty": 4,
                                    "max_difficulty": 16}},
        "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}}})

    seen = []

    def post(batch, difficulty):
        xs, ys = batch
        seen.append(difficulty)
        return xs[:, :difficulty], ys

    eng.set_data_post_process_func(post)
    # pad-collate variable-length rows
    def pad_collate(items):
        xs = torch.nn.utils.rnn.pad_sequence([x for x, _ in items],
                                             batch_first=True)
        ys = torch.tensor([y for _, y in items])
        return xs, ys

    loader = eng.deepspeed_io(DS(), collate_fn=pad_collate)
    eng.set_custom_curriculum_learning_schedule(
        lambda step: min(16, 4 + 2 * step))
    # the sampler advances the schedule itself, one step per batch
    it = iter(loader)
    xs, _ = next(it)
    assert xs.shape[1] <= 4
    xs, _ = next(it)
    assert xs.shape[1] <= 6
    assert seen[0] == 4 and seen[1] == 6, seen


def test_eigenvalue_from_config():
    """eigenvalue config section: engine tags decoder blocks and the
    power iteration returns one positive eigenvalue per block after a
    create_graph backward (MoQ sensitivity input)."""
    from .common import run_local
    run_local(_eigenvalue_worker)


def _eigenvalue_worker(rank=0, world=1):
    import deepspeed_amd

    class Blk(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.fc = torch.nn.Linear(8, 8)

        def forward(self, x):
            return torch.tanh(self.fc(x))

    class M(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.layers = torch.nn.ModuleList([Blk(), Blk()])
            self.out = torch.nn.Linear(8, 1)

        def forward(self, x, labels=None):
            for l in self.layers:
                x = l(x)
            y = self.out(x)
            if labels is not None:
                return torch.nn.functional.mse_loss(y.float(),
                                                    labels.float())
            return y

    eng, _, _, _ = deepspeed_amd.initialize(model=M(), config={
        "train_micro_batch_size_per_gpu": 4,
        "eigenvalue": {"enabled": True, "max_iter": 20, "tol": 1e-2,
                       "layer_name": "layers", "layer_num": 2},
        "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}}})
    assert eng.eigenvalue is not None
    tagged = [m for m in eng.module.modules()
              if getattr(m, "_deepspeed_eigenvalue_block", False)]
    assert len(tagged) == 2
    loss = eng(torch.randn(4, 8).to(eng.device),
               labels=torch.randn(4, 1).to(eng.device))
    loss.backward(create_graph=True)
    evs = eng.eigenvalue.compute_eigenvalue(eng.module)
    assert len(evs) == 2 and all(e >= 0 for e in evs)


def test_save_checkpoint_exclude_frozen(tmp_path):
    """exclude_frozen_parameters (LoRA-style): the model-states file keeps
    only trainable params (reference engine.save_checkpoint kwarg)."""
    from .common import run_local
    run_local(_frozen_worker, args=(str(tmp_path),))


def _frozen_worker(rank, world, tmp):
    import deepspeed_amd

    class M(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.base = torch.nn.Linear(8, 8)
            self.base.weight.requires_grad_(False)
            self.base.bias.requires_grad_(False)
            self.adapter = torch.nn.Linear(8, 1)

        def forward(self, x, labels=None):
            return torch.nn.functional.mse_loss(
                self.adapter(self.base(x)).float(), labels.float())

    eng, _, _, _ = deepspeed_amd.initialize(model=M(), config={
        "train_micro_batch_size_per_gpu": 2,
        "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}}})
    loss = eng(torch.randn(2, 8).to(eng.device),
               labels=torch.randn(2, 1).to(eng.device))
    eng.backward(loss)
    eng.step()
    eng.save_checkpoint(tmp, tag="fz", exclude_frozen_parameters=True)
    import os
    state = torch.load(os.path.join(tmp, "fz",
                                    "mp_rank_00_model_states.pt"),
                       weights_only=False)
    keys = set(state["module"])
    assert "adapter.weight" in keys and "adapter.bias" in keys
    assert not any(k.startswith("base.") for k in keys), keys


def test_load_module_only_and_scheduler_resume(tmp_path):
    """load_module_only=True restores weights but leaves the optimizer
    fresh; a WarmupLR scheduler's position rides the checkpoint."""
    from .common import run_local
    run_local(_module_only_worker, args=(str(tmp_path),))


def _module_only_worker(rank, world, tmp):
    import deepspeed_amd

    class M(torch.nn.Module):
        def __init__(self):
            super().__init__()
            torch.manual_seed(3)
            self.fc = torch.nn.Linear(8, 1)

        def forward(self, x, labels=None):
            return torch.nn.functional.mse_loss(self.fc(x).float(),
                                                labels.float())

    cfg = {"train_micro_batch_size_per_gpu": 2,
           "optimizer": {"type": "AdamW", "params": {"lr": 1e-2}},
           "scheduler": {"type": "WarmupLR",
                         "params": {"warmup_num_steps": 10,
                                    "warmup_max_lr": 1e-2}}}
    eng, _, _, sched = deepspeed_amd.initialize(model=M(), config=cfg)
    for _ in range(4):
        loss = eng(torch.randn(2, 8).to(eng.device),
                   labels=torch.randn(2, 1).to(eng.device))
        eng.backward(loss)
        eng.step()
    lr_after4 = eng.get_lr()[0]
    eng.save_checkpoint(tmp, tag="s4")
    w = eng.module.fc.weight.detach().clone()

    # full resume: scheduler position restored
    eng2, _, _, _ = deepspeed_amd.initialize(model=M(), config=cfg)
    eng2.load_checkpoint(tmp, tag="s4")
    assert abs(eng2.get_lr()[0] - lr_after4) < 1e-9
    torch.testing.assert_close(eng2.module.fc.weight.detach(), w)

    # module-only: weights restored, optimizer state empty
    eng3, _, _, _ = deepspeed_amd.initialize(model=M(), config=cfg)
    eng3.load_checkpoint(tmp, tag="s4", load_module_only=True)
    torch.testing.assert_close(eng3.module.fc.weight.detach(), w)
    inner = getattr(eng3.optimizer, "optimizer", eng3.optimizer)
    assert all(len(s) == 0 or s.get("step", 0) == 0
               for s in inner.state.values()) or not inner.state


def test_random_ltd_with_zero3():
    """random-LTD wrapping composes with ZeRO-3 module-unit partitioning
    (wrap happens before the optimizer builds units)."""
    from .common import run_local
    run_local(_ltd_zero3_worker)


def _ltd_zero3_worker(rank=0, world=1):
    import deepspeed_amd
    from deepspeed_amd.models import LlamaForCausalLM, llama_tiny
    torch.manual_seed(0)
    model = LlamaForCausalLM(llama_tiny())
    engine, _, _, _ = deepspeed_amd.initialize(model=model, config={
        "train_micro_batch_size_per_gpu": 2,
        "bf16": {"enabled": True},
        "zero_optimization": {"stage": 3, "overlap_comm": False},
        "data_efficiency": {
            "enabled": True,
            "random_ltd": {"enabled": True, "layers_attr": "model.layers",
                           "skip_first": 1, "skip_last": 0,
                           "min_value": 8, "max_value": 32,
                           "seq_per_step": 8, "total_ltd_steps": 4}},
        "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}}})
    for _ in range(3):
        ids = torch.randint(0, 500, (2, 32))
        loss = engine(ids.to(engine.device), labels=ids.to(engine.device))
        engine.backward(loss)
        engine.step()
        assert torch.isfinite(loss)
    assert engine.random_ltd_scheduler.current_seq > 8
