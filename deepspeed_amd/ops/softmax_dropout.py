"""Fused masked/alibi softmax + fused dropout wrappers (reference:
inference/csrc/softmax.cu attn_softmax_v2 and transformer
dropout_kernels.cu). CPU falls back to matching torch math so the
semantics are defined everywhere; on GPU these are single HIP kernels."""

from typing import Optional

import torch

from ._loader import get_ext


def fused_softmax(x: torch.Tensor, mask: Optional[torch.Tensor] = None,
                  alibi_slopes: Optional[torch.Tensor] = None,
                  heads: int = 1, scale: float = 1.0,
                  causal: bool = False) -> torch.Tensor:
    """Softmax over the last dim of [..., Sq, Skv]-shaped scores with
    optional additive mask (broadcast per batch), per-head ALiBi slopes
    (bias = slope * (kpos - last_allowed)), and causal truncation."""
    sq = x.size(-2)
    ext = get_ext()
    if ext is not None and x.is_cuda:
        m = mask.contiguous().to(x.dtype) if mask is not None else None
        a = alibi_slopes.float().contiguous() \
            if alibi_slopes is not None else None
        return ext.fused_softmax(x.contiguous(), m, a, heads, sq, scale,
                                 causal)
    # torch reference (same math, fp32 internally)
    n = x.size(-1)
    v = x.float() * scale
    if alibi_slopes is not None:
        q_idx = torch.arange(sq, device=x.device)
        limit = (n - sq) + q_idx + 1 if causal else torch.full_like(q_idx, n)
        rel = torch.arange(n, device=x.device)[None, :] - \
            (limit[:, None] - 1)
        shape = [1] * (x.dim() - 3) + [heads, 1, 1]
        v = v + alibi_slopes.float().view(shape) * rel
    if mask is not None:
        v = v + mask.float()
    if causal:
        q_idx = torch.arange(sq, device=x.device)[:, None]
        k_idx = torch.arange(n, device=x.device)[None, :]
        v = v.masked_fill(k_idx > (n - sq) + q_idx, float("-inf"))
    return torch.softmax(v, dim=-1).to(x.dtype)


class _FusedDropoutFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, bias, residual, ratio, seed):
        ext = get_ext()
        y, mask = ext.fused_dropout(x.contiguous(), bias, residual, ratio,
                                    seed)
        ctx.save_for_backward(mask)
        ctx.ratio = ratio
        ctx.has_bias = bias is not None
        ctx.cols = x.size(-1)
        ctx.has_res = residual is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        (mask,) = ctx.saved_tensors
        ext = get_ext()
        dx = ext.dropout_bwd(dy.contiguous(), mask, ctx.ratio)
        dbias = None
        if ctx.has_bias:
            dbias = dx.reshape(-1, ctx.cols).sum(0)
        dres = dy if ctx.has_res else None
        return dx, dbias, dres, None, None


def fused_bias_dropout_residual(x, bias=None, residual=None, ratio=0.1,
                                seed: Optional[int] = None,
                                training: bool = True):
    """y = dropout(x + bias) + residual, one kernel fwd and bwd on GPU."""
    if not training or ratio == 0.0:
        y = x if bias is None else x + bias
        return y if residual is None else y + residual
    ext = get_ext()
    if ext is None or not x.is_cuda:
        y = x if bias is None else x + bias
        y = torch.nn.functional.dropout(y, p=ratio, training=True)
        return y if residual is None else y + residual
    if seed is None:
        seed = int(torch.randint(0, 2**62, (1,)).item())
    return _FusedDropoutFn.apply(
        x, bias.contiguous() if bias is not None else None,
        residual.contiguous() if residual is not None else None, ratio, seed)
