"""Compression layers — QAT fake-quant + pruning (reference:
deepspeed/compression/basic_layer.py LinearLayer_Compress, utils.py
Quantizer/STE; csrc fake_quantizer.cu is replaced by torch ops feeding the
same groupwise semantics as ops/quantizer.py).
"""

import torch
import torch.nn as nn


class _FakeQuantSTE(torch.autograd.Function):
    """Symmetric per-tensor/per-group fake quant with straight-through
    gradients."""

    @staticmethod
    def forward(ctx, x, bits, group_size):
        qmax = 2.0 ** (bits - 1) - 1
        shape = x.shape
        flat = x.reshape(-1)
        n = flat.numel()
        gs = group_size if group_size > 0 else n
        groups = (n + gs - 1) // gs
        padded = flat
        if groups * gs != n:
            padded = torch.cat([flat, flat.new_zeros(groups * gs - n)])
        g = padded.view(groups, gs)
        scale = (g.abs().amax(dim=1, keepdim=True) / qmax).clamp(min=1e-8)
        q = torch.round(g / scale).clamp(-qmax, qmax) * scale
        return q.view(-1)[:n].view(shape)

    @staticmethod
    def backward(ctx, grad):
        return grad, None, None


def fake_quantize(x, bits=8, group_size=0):
    return _FakeQuantSTE.apply(x, bits, group_size)


class QuantAct(nn.Module):
    """Activation fake-quant with running-range EMA (reference
    basic_layer.py QuantAct)."""

    def __init__(self, bits=8, momentum=0.9):
        super().__init__()
        self.bits = bits
        self.momentum = momentum
        self.register_buffer("range", torch.zeros(1))

    def forward(self, x):
        if self.training:
            cur = x.detach().abs().max().reshape(1)
            self.range.mul_(self.momentum).add_(cur * (1 - self.momentum))
        qmax = 2.0 ** (self.bits - 1) - 1
        scale = (self.range / qmax).clamp(min=1e-8)
        q = torch.round((x / scale).clamp(-qmax, qmax)) * scale
        # straight-through estimator
        return x + (q - x).detach()


class LinearLayer_Compress(nn.Linear):
    """nn.Linear with optional QAT weight quantization, activation
    quantization, magnitude (unstructured) pruning and row (structured)
    pruning — composable, applied at forward time; ``fix_sparsity`` /
    ``fix_weight_quantization`` bake them in."""

    def __init__(self, in_features, out_features, bias=True):
        super().__init__(in_features, out_features, bias=bias)
        self.weight_quant_bits = 0
        self.weight_quant_group = 0
        self.act_quant = None
        self.register_buffer("sparse_mask", torch.ones(0), persistent=False)
        self.register_buffer("row_mask", torch.ones(0), persistent=False)

    # -- enable knobs -------------------------------------------------------
    def enable_weight_quantization(self, bits: int, group_size: int = 0):
        self.weight_quant_bits = bits
        self.weight_quant_group = group_size

    def enable_activation_quantization(self, bits: int):
        self.act_quant = QuantAct(bits)

    def enable_sparse_pruning(self, ratio: float, method: str = "l1"):
        w = self.weight.detach().abs()
        k = int(w.numel() * ratio)
        if k > 0:
            thresh = w.reshape(-1).kthvalue(k).values
            self.sparse_mask = (w > thresh).to(self.weight.dtype)
        else:
            self.sparse_mask = torch.ones_like(self.weight)

    def enable_row_pruning(self, ratio: float):
        norms = self.weight.detach().norm(dim=1)
        k = int(norms.numel() * ratio)
        mask = torch.ones_like(norms)
        if k > 0:
            idx = norms.argsort()[:k]
            mask[idx] = 0.0
        self.row_mask = mask

    # -- forward ------------------------------------------------------------
    def effective_weight(self):
        w = self.weight
        if self.sparse_mask.numel():
            w = w * self.sparse_mask
        if self.row_mask.numel():
            w = w * self.row_mask[:, None]
        if self.weight_quant_bits:
            w = fake_quantize(w, self.weight_quant_bits,
                              self.weight_quant_group)
        return w

    def forward(self, x):
        if self.act_quant is not None:
            x = self.act_quant(x)
        return nn.functional.linear(x, self.effective_weight(), self.bias)

    # -- bake-in ------------------------------------------------------------
    @torch.no_grad()
    def fix_sparsity(self):
        if self.sparse_mask.numel():
            self.weight.mul_(self.sparse_mask)
        if self.row_mask.numel():
            self.weight.mul_(self.row_mask[:, None])

    @torch.no_grad()
    def fix_weight_quantization(self):
        if self.weight_quant_bits:
            self.weight.copy_(fake_quantize(self.weight,
                                            self.weight_quant_bits,
                                            self.weight_quant_group))
            self.weight_quant_bits = 0
