"""Rotary position embedding backed by the CDNA4 HIP kernel.

Kernel parity target: reference apply_rotary_pos_emb.cu (neox rotate-half,
GQA-aware). cos/sin tables are precomputed fp32 on host (per the CDNA4
guide: on-device trig turns this memory-bound op VALU-bound) and cached per
(dim, max_seq, theta, device).
"""

from typing import Optional

import torch

from ._loader import get_ext

_table_cache = {}


def llama3_scale_freqs(inv_freq: torch.Tensor, factor: float = 8.0,
                       low_freq_factor: float = 1.0,
                       high_freq_factor: float = 4.0,
                       original_max_position: int = 8192) -> torch.Tensor:
    """Llama-3.1 rope_scaling (rope_type "llama3"): low-frequency bands
    are divided by `factor`, high-frequency bands untouched, with a
    smooth interpolation between the two wavelength cutoffs (matches HF
    transformers' _compute_llama3_parameters)."""
    low_wl = original_max_position / low_freq_factor
    high_wl = original_max_position / high_freq_factor
    wavelen = 2 * torch.pi / inv_freq
    scaled = torch.where(wavelen > low_wl, inv_freq / factor, inv_freq)
    smooth = (original_max_position / wavelen - low_freq_factor) / (
        high_freq_factor - low_freq_factor)
    smoothed = (1 - smooth) * (inv_freq / factor) + smooth * inv_freq
    mid = (wavelen <= low_wl) & (wavelen >= high_wl)
    return torch.where(mid, smoothed, scaled)


def rope_tables(dim: int, max_seq: int, theta: float = 500000.0,
                device=None, scaling: float = 1.0, rope_scaling=None):
    """fp32 [max_seq, dim/2] cos/sin tables (Llama-3 default theta).
    ``scaling`` divides positions (linear/PI scaling); ``rope_scaling``
    is a llama3-style dict {factor, low_freq_factor, high_freq_factor,
    original_max_position_embeddings} applied to the frequencies."""
    rs_key = tuple(sorted(rope_scaling.items())) if rope_scaling else None
    key = (dim, max_seq, theta, str(device), scaling, rs_key)
    if key not in _table_cache:
        inv_freq = 1.0 / (theta ** (torch.arange(0, dim, 2,
                                                 dtype=torch.float64) / dim))
        if rope_scaling:
            inv_freq = llama3_scale_freqs(
                inv_freq,
                factor=float(rope_scaling.get("factor", 8.0)),
                low_freq_factor=float(
                    rope_scaling.get("low_freq_factor", 1.0)),
                high_freq_factor=float(
                    rope_scaling.get("high_freq_factor", 4.0)),
                original_max_position=int(rope_scaling.get(
                    "original_max_position_embeddings", 8192)))
        t = torch.arange(max_seq, dtype=torch.float64) / scaling
        freqs = torch.outer(t, inv_freq)
        _table_cache[key] = (freqs.cos().float().to(device).contiguous(),
                             freqs.sin().float().to(device).contiguous())
    return _table_cache[key]


def _torch_rope(x, cos, sin, positions=None, backward=False):
    B, S, H, D = x.shape
    half = D // 2
    if positions is not None:
        cos = cos[positions.view(-1).long()].view(B, S, 1, half)
        sin = sin[positions.view(-1).long()].view(B, S, 1, half)
    else:
        cos = cos[:S].view(1, S, 1, half)
        sin = sin[:S].view(1, S, 1, half)
    if backward:
        sin = -sin
    xf = x.float()
    x1, x2 = xf[..., :half], xf[..., half:]
    out = torch.cat([x1 * cos - x2 * sin, x2 * cos + x1 * sin], dim=-1)
    return out.to(x.dtype)


class _RopeFn(torch.autograd.Function):
    """Out-of-place autograd wrapper (kernel itself is in-place on a copy)."""

    @staticmethod
    def forward(ctx, x, cos, sin, positions):
        ctx.save_for_backward(cos, sin,
                              positions if positions is not None else
                              torch.empty(0))
        ext = get_ext()
        if ext is not None and x.is_cuda:
            out = x.contiguous().clone()
            ext.rope(out, cos, sin, positions, False)
            return out
        return _torch_rope(x, cos, sin, positions, backward=False)

    @staticmethod
    def backward(ctx, dy):
        cos, sin, positions = ctx.saved_tensors
        positions = positions if positions.numel() else None
        ext = get_ext()
        if ext is not None and dy.is_cuda:
            dx = dy.contiguous().clone()
            ext.rope(dx, cos, sin, positions, True)
            return dx, None, None, None
        return _torch_rope(dy, cos, sin, positions, backward=True), None, None, None


def apply_rope(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
               positions: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Apply RoPE to [B, S, H, D] (rotate-half pairing i <-> i+D/2)."""
    assert x.dim() == 4, "apply_rope expects [B, S, H, D]"
    return _RopeFn.apply(x, cos, sin, positions)
