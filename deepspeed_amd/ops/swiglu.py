"""Fused SwiGLU / GeGLU activation backed by the CDNA4 HIP kernel.

Kernel parity target: reference gated_activation kernels
(inference/v2/kernels/core_ops/gated_activations) with a training backward.
"""

import torch

from ._loader import get_ext

_SILU, _GELU = 0, 1


class _GatedActFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gate, up, act):
        ctx.act = act
        gate, up = gate.contiguous(), up.contiguous()
        ctx.save_for_backward(gate, up)
        ext = get_ext()
        if ext is not None and gate.is_cuda:
            return ext.gated_act_fwd(gate, up, act)
        gf = gate.float()
        a = torch.nn.functional.silu(gf) if act == _SILU else \
            torch.nn.functional.gelu(gf, approximate="tanh")
        return (a * up.float()).to(gate.dtype)

    @staticmethod
    def backward(ctx, dout):
        gate, up = ctx.saved_tensors
        ext = get_ext()
        dout = dout.contiguous()
        if ext is not None and gate.is_cuda:
            dgate, dup = ext.gated_act_bwd(dout, gate, up, ctx.act)
            return dgate, dup, None
        gf, uf, dof = gate.float(), up.float(), dout.float()
        if ctx.act == _SILU:
            sig = torch.sigmoid(gf)
            a = gf * sig
            da = sig * (1 + gf * (1 - sig))
        else:
            k = 0.7978845608028654
            t = torch.tanh(k * (gf + 0.044715 * gf ** 3))
            a = 0.5 * gf * (1 + t)
            da = 0.5 * (1 + t) + 0.5 * gf * (1 - t * t) * k * (
                1 + 3 * 0.044715 * gf * gf)
        dgate = (dof * uf * da).to(gate.dtype)
        dup = (dof * a).to(up.dtype)
        return dgate, dup, None


def swiglu(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    """silu(gate) * up, fused."""
    return _GatedActFn.apply(gate, up, _SILU)


def geglu(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    """gelu(gate) * up, fused."""
    return _GatedActFn.apply(gate, up, _GELU)
