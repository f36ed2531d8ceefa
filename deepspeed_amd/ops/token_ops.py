"""Token gather/scatter ops for random-LTD (reference:
csrc/random_ltd/gather_scatter.cu + deepspeed/runtime/data_pipeline/
data_routing/basic_layer.py). Row-coalesced HIP kernels on GPU; torch
gather/scatter fallback elsewhere. Indices must select DISTINCT tokens
per batch row (random-LTD samples without replacement), so scatter is a
plain overwrite and backward needs no atomics."""

import torch

from ._loader import get_ext


def _kernel_ok(x: torch.Tensor) -> bool:
    return (get_ext() is not None and x.is_cuda
            and x.dtype in (torch.bfloat16, torch.float16, torch.float32)
            and x.size(-1) % (16 // x.element_size()) == 0)


class _TokenGatherFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, idx):
        ctx.save_for_backward(idx)
        ctx.S = x.size(1)
        if _kernel_ok(x):
            return get_ext().token_gather(x.contiguous(), idx.int())
        g = idx.long().unsqueeze(-1).expand(-1, -1, x.size(-1))
        return x.gather(1, g)

    @staticmethod
    def backward(ctx, dout):
        (idx,) = ctx.saved_tensors
        B, K, D = dout.shape
        base = dout.new_zeros(B, ctx.S, D)
        if _kernel_ok(dout):
            dx = get_ext().token_scatter(base, dout.contiguous(), idx.int())
        else:
            g = idx.long().unsqueeze(-1).expand(-1, -1, D)
            dx = base.scatter(1, g, dout)
        return dx, None


class _TokenScatterFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, base, sub, idx):
        ctx.save_for_backward(idx)
        if _kernel_ok(base):
            return get_ext().token_scatter(base.contiguous(),
                                           sub.contiguous(), idx.int())
        g = idx.long().unsqueeze(-1).expand(-1, -1, base.size(-1))
        return base.scatter(1, g, sub)

    @staticmethod
    def backward(ctx, dout):
        (idx,) = ctx.saved_tensors
        g = idx.long().unsqueeze(-1).expand(-1, -1, dout.size(-1))
        if _kernel_ok(dout):
            dsub = get_ext().token_gather(dout.contiguous(), idx.int())
        else:
            dsub = dout.gather(1, g)
        dbase = dout.scatter(1, g, torch.zeros_like(dsub))
        return dbase, dsub, None


def token_gather(x: torch.Tensor, idx: torch.Tensor) -> torch.Tensor:
    """x [B,S,D], idx [B,K] (distinct per row) -> [B,K,D]."""
    return _TokenGatherFn.apply(x, idx)


def token_scatter(base: torch.Tensor, sub: torch.Tensor,
                  idx: torch.Tensor) -> torch.Tensor:
    """Rows of ``base`` at ``idx`` replaced by ``sub`` (out-of-place)."""
    return _TokenScatterFn.apply(base, sub, idx)
