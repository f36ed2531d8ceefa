"""Flash-attention forward (hand-written CDNA4 MFMA kernel).

See csrc/attention.hip for the kernel design (probe-verified MFMA
layouts, swapped-QK^T online softmax). This wrapper owns the V transpose
(the kernel consumes vt[B,Hkv,D,S] so its PV fragments are contiguous
16-byte loads) and the eligibility check, falling back to torch SDPA where
the kernel does not apply (training backward, D!=128, ragged seq, CPU).
"""

import math
from typing import Optional

import torch

from ._loader import get_ext


def _eligible(q: torch.Tensor, k: torch.Tensor) -> bool:
    return (get_ext() is not None and q.is_cuda
            and q.dtype == torch.bfloat16
            and q.size(-1) == 128 and q.size(1) % 32 == 0
            and q.size(1) == k.size(1))


def flash_attn_available(q: torch.Tensor, k: torch.Tensor) -> bool:
    """Inference/prefill forward: ON by default (v3 kernel beats aotriton
    SDPA on the bench shape, 417 vs <=402 TF). DS_AMD_FLASH=0 opts out."""
    import os
    if os.environ.get("DS_AMD_FLASH", "1") == "0":
        return False
    return _eligible(q, k) and not torch.is_grad_enabled()


def flash_train_available(q: torch.Tensor, k: torch.Tensor) -> bool:
    """Training fwd+bwd path (FlashAttnFunc). Gated by DS_AMD_FLASH_TRAIN
    until the v2 backward beats the aotriton backward on the bench shape
    (flip the default then)."""
    import os
    if os.environ.get("DS_AMD_FLASH_TRAIN", "0") != "1":
        return False
    return _eligible(q, k)


def _vt_from_bshd(v):
    """[B, S, Hkv, D] -> [B, Hkv, D, S] via the tiled transpose kernel."""
    ext = get_ext()
    B, S, Hk, D = v.shape
    if ext is None or not v.is_cuda or v.dtype != torch.bfloat16:
        return v.permute(0, 2, 3, 1).contiguous()
    out = ext.transpose_bf16(v.contiguous(), B * Hk, S, D, Hk * D, Hk, D,
                             S * Hk * D)
    return out.view(B, Hk, D, S)


def _t_last2_bhsd(x):
    """[B, H, S, D] -> [B, H, D, S] via the tiled transpose kernel."""
    ext = get_ext()
    B, H, S, D = x.shape
    if ext is None or not x.is_cuda or x.dtype != torch.bfloat16:
        return x.transpose(-1, -2).contiguous()
    out = ext.transpose_bf16(x.contiguous(), B * H, S, D, D, 1, 0, S * D)
    return out.view(B, H, D, S)


def flash_attn_fwd(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                   causal: bool = True,
                   scale: Optional[float] = None) -> torch.Tensor:
    """q,k,v: [B, S, H(kv), D] bf16 -> o [B, S, H, D]. Inference/prefill
    only (no autograd)."""
    ext = get_ext()
    scale = scale if scale is not None else 1.0 / math.sqrt(q.size(-1))
    vt = _vt_from_bshd(v)  # [B, Hkv, D, S]
    return ext.flash_attn_fwd(q.contiguous(), k.contiguous(), vt, scale,
                              causal)


def sdpa_reference(q, k, v, causal=True, scale=None):
    """Plain fp32 reference in the same [B,S,H,D] layout (for tests)."""
    qt, kt, vt = (t.transpose(1, 2).float() for t in (q, k, v))
    rep = qt.shape[1] // kt.shape[1]
    if rep > 1:
        kt = kt.repeat_interleave(rep, dim=1)
        vt = vt.repeat_interleave(rep, dim=1)
    o = torch.nn.functional.scaled_dot_product_attention(
        qt, kt, vt, is_causal=causal, scale=scale)
    return o.transpose(1, 2).to(q.dtype)


def flash_attn_bwd(q, k, v, o, do, lse, causal=True, scale=None):
    """Backward pass in BHSD ([B,H(kv),S,D] bf16; lse [B,H,S] fp32).
    Returns (dq, dk, dv). GPU-validation pending (round-2 item) — the
    tile algebra is CPU-verified in ops/flash_bwd_ref.py."""
    ext = get_ext()
    scale = scale if scale is not None else 1.0 / math.sqrt(q.size(-1))
    delta = (do.float() * o.float()).sum(-1)              # [B,H,S]
    qt = _t_last2_bhsd(q)
    kt = _t_last2_bhsd(k)
    dot = _t_last2_bhsd(do)
    return ext.flash_attn_bwd(q.contiguous(), k.contiguous(), v.contiguous(),
                              do.contiguous(), qt, kt, dot,
                              lse.contiguous().float(), delta.contiguous(),
                              scale, causal)


class FlashAttnFunc(torch.autograd.Function):
    """Training-path flash attention (BSHD in/out). GPU-validation of the
    backward kernels is a round-2 gate; until then the model keeps SDPA."""

    @staticmethod
    def forward(ctx, q, k, v, causal, scale):
        ext = get_ext()
        scale = scale if scale is not None else 1.0 / math.sqrt(q.size(-1))
        vt = _vt_from_bshd(v)
        o, lse = ext.flash_attn_fwd_lse(q.contiguous(), k.contiguous(), vt,
                                        scale, causal)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.causal, ctx.scale = causal, scale
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse = ctx.saved_tensors
        # kernels run in BHSD
        qh, kh, vh, oh, doh = (t.transpose(1, 2).contiguous()
                               for t in (q, k, v, o, do))
        dq, dk, dv = flash_attn_bwd(qh, kh, vh, oh, doh, lse,
                                    causal=ctx.causal, scale=ctx.scale)
        return (dq.transpose(1, 2), dk.transpose(1, 2),
                dv.transpose(1, 2), None, None)


def flash_attn_func(q, k, v, causal=True, scale=None):
    """Differentiable flash attention, [B,S,H(kv),D] bf16."""
    return FlashAttnFunc.apply(q, k, v, causal, scale)
