"""Fused cross-entropy (bf16 logits, no fp32 materialization).

See csrc/cross_entropy.hip. The autograd wrapper recomputes nothing: the
forward saves (logits, labels, lse) and the backward emits bf16 dlogits in
one elementwise pass. Enabled in the chunked-CE loss path with
DS_AMD_FUSED_CE=1 (GPU numerics validation is a round-2 gate; the torch
fallback defines the semantics and runs everywhere)."""

import os

import torch

from ._loader import get_ext


class _FusedCE(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, labels, ignore_index):
        ext = get_ext()
        loss, lse = ext.ce_fwd(logits, labels, ignore_index)
        ctx.save_for_backward(logits, labels, lse)
        ctx.ignore_index = ignore_index
        count = (labels != ignore_index).sum().float()
        ctx.mark_non_differentiable(count)
        return loss.sum(), count

    @staticmethod
    def backward(ctx, gsum, _gcount):
        logits, labels, lse = ctx.saved_tensors
        gscale = gsum.expand(logits.size(0)).contiguous().float()
        d = get_ext().ce_bwd(logits, labels, lse, gscale, ctx.ignore_index)
        return d, None, None


def fused_ce_available(logits) -> bool:
    """Default ON (GPU-numerics-validated round 2, r2_call3): the fused
    kernel computes loss+lse from bf16 logits and emits bf16 dlogits in
    one pass — no [tokens, vocab] fp32 materialization, no cunn_SoftMax
    pair. DS_AMD_FUSED_CE=0 opts out."""
    return (os.environ.get("DS_AMD_FUSED_CE", "1") != "0"
            and get_ext() is not None and logits.is_cuda
            and logits.dtype == torch.bfloat16)


def fused_cross_entropy_sum(logits, labels, ignore_index=-100):
    """Returns (summed loss, valid-token count) like the chunked-CE inner
    loop expects. Falls back to the fp32 torch path when ineligible."""
    if fused_ce_available(logits):
        return _FusedCE.apply(logits.contiguous(), labels.contiguous(),
                              ignore_index)
    lf = logits.float()
    loss = torch.nn.functional.cross_entropy(lf, labels,
                                             ignore_index=ignore_index,
                                             reduction="sum")
    return loss, (labels != ignore_index).sum().float()
