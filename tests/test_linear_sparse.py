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


def test_block_sparse_gather_matches_masked_sdpa():
    """block_sparse_attention (gather path, only live blocks computed) must
    equal SDPA over the block-expanded mask, for Fixed (incl. causal),
    BigBird, and a random layout."""
    import torch.nn.functional as F
    from deepspeed_amd.ops.sparse_attention import (
        BigBirdSparsityConfig, FixedSparsityConfig, block_sparse_attention)
    torch.manual_seed(0)
    B, H, S, D, bs = 2, 2, 64, 16, 16
    q, k, v = (torch.randn(B, H, S, D) for _ in range(3))
    layouts = [
        FixedSparsityConfig(H, block=bs, num_local_blocks=2).make_layout(S),
        FixedSparsityConfig(H, block=bs, num_local_blocks=2,
                            attention="unidirectional").make_layout(S),
        BigBirdSparsityConfig(H, block=bs, num_random_blocks=1,
                              num_sliding_window_blocks=1).make_layout(S),
        torch.rand(H, S // bs, S // bs) < 0.4,
    ]
    for layout in layouts:
        # every query block must see >= 1 key block (SDPA NaNs otherwise)
        layout[..., 0] |= layout.sum(-1) == 0
        got = block_sparse_attention(q, k, v, layout, bs)
        mask = layout.repeat_interleave(bs, 1).repeat_interleave(bs, 2)
        want = F.scaled_dot_product_attention(q, k, v,
                                              attn_mask=mask.unsqueeze(0))
        torch.testing.assert_close(got, want, atol=2e-5, rtol=1e-4)


def test_sparse_self_attention_default_uses_gather_path():
    from deepspeed_amd.ops.sparse_attention import (FixedSparsityConfig,
                                                    SparseSelfAttention)
    torch.manual_seed(1)
    B, H, S, D = 1, 2, 64, 16
    q, k, v = (torch.randn(B, H, S, D) for _ in range(3))
    att = SparseSelfAttention(FixedSparsityConfig(H, block=16,
                                                  num_local_blocks=2))
    out_gather = att(q, k, v)
    out_masked = att(q, k, v, attn_mask=torch.ones(S, S, dtype=torch.bool))
    torch.testing.assert_close(out_gather, out_masked, atol=2e-5, rtol=1e-4)


def test_sparse_attention_config_block():
    """ds_config "sparse_attention" block -> SparseSelfAttention factory
    (reference top-level config schema)."""
    from deepspeed_amd.config import Config
    from deepspeed_amd.ops.sparse_attention import (build_sparse_attention,
                                                    FixedSparsityConfig)
    cfg = Config({"train_micro_batch_size_per_gpu": 1,
                  "sparse_attention": {"mode": "fixed", "block": 16,
                                       "num_local_blocks": 2,
                                       "attention": "unidirectional"}})
    att = build_sparse_attention(cfg.sparse_attention, num_heads=2)
    assert isinstance(att.config, FixedSparsityConfig)
    assert att.config.block == 16 and att.config.causal
    q = torch.randn(1, 2, 64, 8)
    out = att(q, q, q)
    assert out.shape == q.shape


def test_block_sparse_attention_backward():
    """The gather path must be trainable: grads match the masked-SDPA
    reference."""
    import torch.nn.functional as F
    from deepspeed_amd.ops.sparse_attention import (FixedSparsityConfig,
                                                    block_sparse_attention)
    torch.manual_seed(2)
    B, H, S, D, bs = 1, 2, 32, 8, 16
    layout = FixedSparsityConfig(H, block=bs,
                                 num_local_blocks=1).make_layout(S)
    q1, k1, v1 = (torch.randn(B, H, S, D, requires_grad=True)
                  for _ in range(3))
    q2 = q1.detach().clone().requires_grad_()
    k2 = k1.detach().clone().requires_grad_()
    v2 = v1.detach().clone().requires_grad_()
    out1 = block_sparse_attention(q1, k1, v1, layout, bs)
    mask = layout.repeat_interleave(bs, 1).repeat_interleave(bs, 2)
    out2 = F.scaled_dot_product_attention(q2, k2, v2,
                                          attn_mask=mask.unsqueeze(0))
    g = torch.randn_like(out1)
    out1.backward(g)
    out2.backward(g)
    for a, b in ((q1, q2), (k1, k2), (v1, v2)):
        torch.testing.assert_close(a.grad, b.grad, atol=2e-5, rtol=1e-4)


def test_longformer_and_variable_layouts():
    """BSLongformer and Variable sparsity configs (reference
    sparsity_config.py): window/global/varying-window semantics, and both
    run through the gather compute path."""
    from deepspeed_amd.ops.sparse_attention import (
        BSLongformerSparsityConfig, VariableSparsityConfig,
        SparseSelfAttention, block_sparse_attention)
    import torch.nn.functional as F
    H, S, bs = 2, 96, 16
    lf = BSLongformerSparsityConfig(H, block=bs,
                                    num_sliding_window_blocks=3,
                                    global_block_indices=(0,))
    L = lf.make_layout(S)
    n = S // bs
    assert L[:, 0, :].all() and L[:, :, 0].all()        # global row+col
    assert L[0, 3, 2] and L[0, 3, 4] and not L[0, 3, 5]  # window of 3
    var = VariableSparsityConfig(H, block=bs, local_window_blocks=(1, 2),
                                 attention="unidirectional")
    V = var.make_layout(S)
    assert V[0, 1, 1] and V[0, 1, 2].logical_not()       # window [1,3) causal
    assert V[0, 2, 1] and not V[0, 2, 3]
    assert torch.equal(V[0], V[0].tril())                # causal
    torch.manual_seed(3)
    q = torch.randn(1, H, S, 8)
    for layout in (L, V):
        got = block_sparse_attention(q, q, q, layout, bs)
        mask = layout.repeat_interleave(bs, 1).repeat_interleave(bs, 2)
        want = F.scaled_dot_product_attention(q, q, q,
                                              attn_mask=mask.unsqueeze(0))
        torch.testing.assert_close(got, want, atol=2e-5, rtol=1e-4)


def test_sparse_self_attention_module():
    """SparseSelfAttention module (reference sparse_self_attention.py):
    layout-cached forward equals dense SDPA under the block mask; key
    padding masks fold in exactly; pad/unpad helpers round-trip."""
    import math
    from deepspeed_amd.ops.sparse_attention import (
        SparseSelfAttention, FixedSparsityConfig, SparseAttentionUtils,
        layout_to_dense_mask)
    torch.manual_seed(0)
    B, H, S, D = 2, 4, 64, 16
    cfg = FixedSparsityConfig(num_heads=H, block=16)
    attn = SparseSelfAttention(cfg)
    q, k, v = (torch.randn(B, H, S, D) for _ in range(3))
    out = attn(q, k, v)
    mask = layout_to_dense_mask(cfg.make_layout(S), 16)
    scores = (q.float() @ k.float().transpose(-1, -2)) / math.sqrt(D)
    scores = scores.masked_fill(~mask[None], float("-inf"))
    ref = (torch.softmax(scores, -1) @ v.float()).to(q.dtype)
    torch.testing.assert_close(out, ref, rtol=2e-3, atol=2e-3)
    kpm = torch.ones(B, S)
    kpm[:, 50:] = 0
    out2 = attn(q, k, v, key_padding_mask=kpm)
    scores2 = scores.masked_fill(~kpm.bool()[:, None, None, :],
                                 float("-inf"))
    ref2 = torch.softmax(scores2, -1) @ v.float()
    torch.testing.assert_close(out2.float(), ref2, rtol=2e-3, atol=2e-3)
    pad, ids, am = SparseAttentionUtils.pad_to_block_size(
        16, torch.ones(2, 60, dtype=torch.long), torch.ones(2, 60))
    assert pad == 4 and ids.shape[1] == 64 and am.shape[1] == 64
    assert SparseAttentionUtils.unpad_sequence_output(
        pad, torch.zeros(2, 64, 8)).shape[1] == 60
