"""Trainable tensor-parallel linear layers (reference:
deepspeed/module_inject/layers.py LinearLayer autograd :370 /
LinearAllreduce :300 and runtime/tensor_parallel/).

Megatron-style f/g conjugate pairs over the TP group:
  ColumnParallelLinear : y_local = x W_colᵀ       (identity fwd / all-reduce
                         of dx in bwd — the input is logically replicated)
  RowParallelLinear    : y = Σ_tp x_local W_rowᵀ  (all-reduce fwd / identity
                         bwd)
Replicated modules between a row output and a column input see identical
activations AND identical gradients on every TP rank, so their parameters
need no extra synchronization — the invariant the tests assert.
"""

import torch
import torch.nn as nn

from ... import comm as dist


class _CopyToTP(torch.autograd.Function):
    """Identity forward; all-reduce gradient over the TP group ("f")."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return x

    @staticmethod
    def backward(ctx, grad):
        if dist.get_world_size(ctx.group) > 1:
            grad = grad.contiguous()
            dist.all_reduce(grad, group=ctx.group)
        return grad, None


class _ReduceFromTP(torch.autograd.Function):
    """All-reduce forward; identity gradient ("g")."""

    @staticmethod
    def forward(ctx, x, group):
        if dist.get_world_size(group) > 1:
            x = x.contiguous()
            dist.all_reduce(x, group=group)
        return x

    @staticmethod
    def backward(ctx, grad):
        return grad, None


class _ColumnParallelFn(torch.autograd.Function):
    """Fused column-parallel linear with the Domino/Megatron backward
    overlap (reference runtime/domino/transformer.py): the dx all-reduce
    launches ASYNC and the weight/bias gradient GEMMs compute under it —
    the collective rides xGMI while MFMA does the dW math."""

    @staticmethod
    def forward(ctx, x, weight, bias, group):
        ctx.save_for_backward(x, weight)
        ctx.group = group
        ctx.has_bias = bias is not None
        return nn.functional.linear(x, weight, bias)

    @staticmethod
    def backward(ctx, grad):
        x, weight = ctx.saved_tensors
        grad = grad.contiguous()
        dx = grad @ weight
        handle = None
        if dist.get_world_size(ctx.group) > 1:
            handle = dist.all_reduce(dx, group=ctx.group, async_op=True)
        g2 = grad.reshape(-1, grad.shape[-1])
        x2 = x.reshape(-1, x.shape[-1])
        dw = g2.t() @ x2                       # overlaps the dx all-reduce
        db = g2.sum(0) if ctx.has_bias else None
        if handle is not None:
            handle.wait()
        return dx, dw, db, None


class ColumnParallelLinear(nn.Module):
    def __init__(self, weight_shard: torch.Tensor, bias_shard, group):
        super().__init__()
        self.weight = nn.Parameter(weight_shard)
        self.bias = nn.Parameter(bias_shard) if bias_shard is not None else None
        self.group = group
        self.weight.tensor_model_parallel = True
        if self.bias is not None:
            self.bias.tensor_model_parallel = True

    def forward(self, x):
        return _ColumnParallelFn.apply(x, self.weight, self.bias, self.group)


class RowParallelLinear(nn.Module):
    def __init__(self, weight_shard: torch.Tensor, bias_full, group):
        super().__init__()
        self.weight = nn.Parameter(weight_shard)
        # bias applied once after the reduction; kept replicated
        self.bias = nn.Parameter(bias_full) if bias_full is not None else None
        self.group = group
        self.weight.tensor_model_parallel = True

    def forward(self, x):
        y = nn.functional.linear(x, self.weight)
        y = _ReduceFromTP.apply(y, self.group)
        if self.bias is not None:
            y = y + self.bias
        return y


@torch.no_grad()
def shard_linear_for_training(model: nn.Module, tp_group, tp_rank: int,
                              tp_size: int):
    """In-place replace nn.Linear with trainable TP shards by the same name
    policy as inference AutoTP (see inference/auto_tp.py patterns)."""
    from ...inference.auto_tp import (COLUMN_PATTERNS, ROW_PATTERNS, _match,
                                      shard_attention_heads)
    replaced = 0
    for parent_name, parent in list(model.named_modules()):
        for child_name, child in list(parent._modules.items()):
            if not isinstance(child, nn.Linear):
                continue
            full = f"{parent_name}.{child_name}" if parent_name else child_name
            W = child.weight.data
            b = child.bias.data if child.bias is not None else None
            if _match(full, COLUMN_PATTERNS):
                out = W.size(0)
                assert out % tp_size == 0, f"{full}: {out} % {tp_size}"
                sl = slice(tp_rank * out // tp_size,
                           (tp_rank + 1) * out // tp_size)
                new = ColumnParallelLinear(
                    W[sl].clone(), b[sl].clone() if b is not None else None,
                    tp_group)
            elif _match(full, ROW_PATTERNS):
                inp = W.size(1)
                assert inp % tp_size == 0, f"{full}: {inp} % {tp_size}"
                sl = slice(tp_rank * inp // tp_size,
                           (tp_rank + 1) * inp // tp_size)
                new = RowParallelLinear(
                    W[:, sl].clone(), b.clone() if b is not None else None,
                    tp_group)
            else:
                continue
            parent._modules[child_name] = new
            replaced += 1
    shard_attention_heads(model, tp_rank, tp_size)
    return replaced
