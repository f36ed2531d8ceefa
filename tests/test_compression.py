"""Compression tests (reference contract:
tests/unit/compression/test_compression.py): QAT fake-quant STE, pruning
masks, init_compression replacement + redundancy_clean bake-in."""

import torch
import torch.nn as nn


def test_fake_quantize_ste():
    from deepspeed_amd.compression import fake_quantize
    x = torch.randn(100, requires_grad=True)
    y = fake_quantize(x, bits=4)
    # quantized values on a 4-bit grid per-tensor
    assert y.unique().numel() <= 15
    g = torch.randn(100)
    y.backward(g)
    assert torch.equal(x.grad, g)  # straight-through


def test_linear_compress_pruning():
    from deepspeed_amd.compression import LinearLayer_Compress
    torch.manual_seed(0)
    lin = LinearLayer_Compress(16, 8)
    lin.enable_sparse_pruning(0.5)
    w = lin.effective_weight()
    assert ((w == 0).float().mean() - 0.5).abs() < 0.05
    lin.enable_row_pruning(0.25)
    w = lin.effective_weight()
    assert (w.norm(dim=1) == 0).sum() >= 2
    lin.fix_sparsity()
    assert ((lin.weight == 0).float().mean()) > 0.5


def test_init_compression_and_clean():
    from deepspeed_amd.compression import (LinearLayer_Compress,
                                           init_compression,
                                           redundancy_clean)
    torch.manual_seed(1)
    model = nn.Sequential(nn.Linear(8, 8), nn.ReLU(), nn.Linear(8, 4))
    cfg = {"weight_quantization": {"different_groups": {
        "g": {"params": {"target_bits": 8}, "modules": ["^0$", "^2$"]}}},
        "sparse_pruning": {"different_groups": {
            "s": {"params": {"dense_ratio": 0.5}, "modules": ["^0$"]}}}}
    init_compression(model, cfg)
    assert isinstance(model[0], LinearLayer_Compress)
    assert isinstance(model[2], LinearLayer_Compress)
    assert model[0].weight_quant_bits == 8
    assert model[0].sparse_mask.numel() > 0 and model[2].sparse_mask.numel() == 0

    x = torch.randn(3, 8)
    y1 = model(x)
    redundancy_clean(model)
    y2 = model(x)
    torch.testing.assert_close(y1, y2, rtol=1e-5, atol=1e-6)
    # training still works through STE
    model(x).sum().backward()
    assert model[0].weight.grad is not None


def test_quant_act_ema():
    from deepspeed_amd.compression import QuantAct
    qa = QuantAct(bits=8)
    qa.train()
    x = torch.randn(1000) * 3
    for _ in range(20):
        qa(x)
    assert qa.range.item() > 1.0
    qa.eval()
    y = qa(x)
    # inside the learned range the error is below one quant step; values
    # beyond it clamp (standard QAT behavior)
    inside = x.abs() <= qa.range.item()
    assert (y - x)[inside].abs().max() < 0.1
