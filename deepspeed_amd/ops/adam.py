"""FusedAdam optimizer (HIP flat kernel) + the ZeRO flat-shard step hook.

Capability parity with the reference's FusedAdam
(deepspeed/ops/adam/fused_adam.py:18 over csrc/adam/multi_tensor_adam.cu).
MI355X-first difference: the ZeRO optimizers in this framework already hold
params/grads/state as contiguous flat shards, so the hot step is ONE
grid-stride HIP kernel per bucket (optionally fusing the fp32->bf16 param
write) instead of a multi-tensor-apply chunk harness.
"""

import math
from typing import Optional

import torch

from ._loader import get_ext, has_ext


class FusedAdam(torch.optim.Optimizer):
    """Adam/AdamW. On GPU each param steps through the HIP fused kernel
    (fp32 master held in state for 16-bit params); on CPU falls back to a
    vectorized torch implementation."""

    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0.0, adam_w_mode=True, bias_correction=True,
                 amsgrad=False, set_grad_none=True):
        if amsgrad:
            raise ValueError("amsgrad not supported")
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay,
                        bias_correction=bias_correction)
        super().__init__(params, defaults)
        self.adam_w_mode = adam_w_mode
        self.set_grad_none = set_grad_none

    def zero_grad(self, set_to_none: Optional[bool] = None):
        if set_to_none is None:
            set_to_none = self.set_grad_none
        super().zero_grad(set_to_none=set_to_none)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        ext = get_ext()
        for group in self.param_groups:
            lr = group["lr"]
            beta1, beta2 = group["betas"]
            eps = group["eps"]
            wd = group["weight_decay"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    master = p if p.dtype == torch.float32 else p.float()
                    if p.dtype != torch.float32:
                        state["master"] = master.detach().clone()
                    state["exp_avg"] = torch.zeros_like(master,
                                                        dtype=torch.float32)
                    state["exp_avg_sq"] = torch.zeros_like(master,
                                                           dtype=torch.float32)
                state["step"] += 1
                master = state.get("master", p)
                if ext is not None and p.is_cuda:
                    p16 = p.view(-1) if p.dtype == torch.bfloat16 else None
                    ext.fused_adam_flat(
                        master.view(-1), p.grad.contiguous().view(-1),
                        state["exp_avg"].view(-1), state["exp_avg_sq"].view(-1),
                        p16, lr, beta1, beta2, eps, wd, state["step"], 1.0,
                        self.adam_w_mode)
                    if p.dtype not in (torch.bfloat16, torch.float32):
                        p.copy_(master)
                else:
                    _torch_adam_step(master, p.grad, state["exp_avg"],
                                     state["exp_avg_sq"], lr, beta1, beta2, eps,
                                     wd, state["step"], self.adam_w_mode)
                    if master is not p:
                        p.copy_(master)
        return loss


@torch.no_grad()
def _torch_adam_step(p, grad, m, v, lr, beta1, beta2, eps, wd, step, adamw,
                     inv_scale=1.0):
    g = grad.float()
    if inv_scale != 1.0:
        g = g * inv_scale
    if not adamw and wd != 0.0:
        g = g + wd * p
    m.mul_(beta1).add_(g, alpha=1 - beta1)
    v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
    bc1 = 1 - beta1 ** step
    bc2_sqrt = math.sqrt(1 - beta2 ** step)
    denom = v.sqrt().div(bc2_sqrt).add_(eps)
    update = (m / bc1).div_(denom)
    if adamw and wd != 0.0:
        update = update.add(p, alpha=wd)
    p.add_(update, alpha=-lr)


def multi_tensor_adam_available() -> bool:
    return has_ext() or not torch.cuda.is_available()


_ADAM_TYPES = (FusedAdam, torch.optim.Adam, torch.optim.AdamW)


def _adam_hyperparams(optimizer, group):
    beta1, beta2 = group["betas"]
    adamw = (getattr(optimizer, "adam_w_mode", None)
             if isinstance(optimizer, FusedAdam)
             else isinstance(optimizer, torch.optim.AdamW))
    return group["lr"], beta1, beta2, group["eps"], group["weight_decay"], adamw


def fused_adam_step(optimizer, group, master_param, grad_shard, combined_scale,
                    segments=None, bump_step=True) -> bool:
    """ZeRO flat-shard step: fused HIP kernel over the group's fp32 master
    shard with 16-bit grads consumed in place.

    ``segments`` (optional) is a list of ``(offset, numel, out16_or_None)``
    or ``(offset, numel, out16_or_None, grad_tensor)`` covering the shard;
    when given, the kernel runs per segment and fuses the fp32->16-bit
    param-shard write (out16 is the bucket's shard view), saving a full
    extra read+write pass over the shard. A 4th element overrides the grad
    source for that segment (ZeRO-3 direct-grad mode feeds the 16-bit
    reduce-scatter output straight in — ``grad_shard`` may then be empty).
    Returns False if this optimizer cannot be fused (caller falls back to
    torch).
    """
    if not isinstance(optimizer, _ADAM_TYPES):
        return False
    lr, beta1, beta2, eps, wd, adamw = _adam_hyperparams(optimizer, group)
    state = optimizer.state.setdefault(master_param, {})
    if "exp_avg" not in state:
        state["step"] = 0
        state["exp_avg"] = torch.zeros_like(master_param, dtype=torch.float32)
        state["exp_avg_sq"] = torch.zeros_like(master_param, dtype=torch.float32)
    if bump_step:
        state["step"] += 1
    scale = float(combined_scale) if not torch.is_tensor(combined_scale) \
        else float(combined_scale.item())
    inv_scale = 1.0 / scale
    if segments is None:
        segments = [(0, master_param.numel(), None)]
    ext = get_ext()
    master = master_param.data.view(-1)
    grads = grad_shard.view(-1)
    m = state["exp_avg"].view(-1)
    v = state["exp_avg_sq"].view(-1)
    use_cpu_ext = (ext is not None and not master_param.is_cuda
                   and hasattr(ext, "cpu_adam_flat"))
    host16 = None
    if use_cpu_ext and any(len(s) > 2 and s[2] is not None and s[2].is_cuda
                           for s in segments):
        # ZeRO-Offload: the host Adam writes updated params as bf16 into a
        # cached pinned buffer; the H2D ships half the bytes of an fp32
        # master copy and needs no GPU cast kernel.
        host16 = state.get("host_bf16")
        if host16 is None or host16.numel() != master.numel():
            host16 = torch.empty(master.numel(), dtype=torch.bfloat16)
            if torch.cuda.is_available():
                host16 = host16.pin_memory()
            state["host_bf16"] = host16
    for seg in segments:
        off, n, out16 = seg[0], seg[1], seg[2]
        g = seg[3].view(-1) if len(seg) > 3 else grads[off:off + n]
        sl = slice(off, off + n)
        if out16 is not None and out16.dtype != torch.bfloat16:
            out16 = None  # kernel only fuses bf16 writes; caller copies
        if ext is not None and master_param.is_cuda:
            ext.fused_adam_flat(master[sl], g, m[sl], v[sl],
                                out16.view(-1) if out16 is not None else None,
                                lr, beta1, beta2, eps, wd, state["step"],
                                inv_scale, adamw)
        elif use_cpu_ext and not g.is_cuda:
            w16 = None
            if out16 is not None:
                w16 = host16[sl] if out16.is_cuda else out16.view(-1)
            ext.cpu_adam_flat(master[sl], g, m[sl], v[sl], w16, lr, beta1,
                              beta2, eps, wd, state["step"], inv_scale, adamw)
            if out16 is not None and out16.is_cuda:
                out16.view(-1).copy_(host16[sl], non_blocking=True)
        else:
            _torch_adam_step(master[sl], g, m[sl], v[sl], lr, beta1,
                             beta2, eps, wd, state["step"], adamw, inv_scale)
            if out16 is not None:
                out16.view(-1).copy_(master[sl])
    return True


class DeepSpeedCPUAdam(FusedAdam):
    """Host-resident Adam stepped by the AVX kernel (reference
    ops/adam/cpu_adam.py DeepSpeedCPUAdam — the user-facing optimizer for
    hand-rolled ZeRO-Offload setups). Params must live on CPU; 16-bit
    params keep an fp32 master in state like FusedAdam."""

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        ext = get_ext()
        for group in self.param_groups:
            lr = group["lr"]
            beta1, beta2 = group["betas"]
            eps = group["eps"]
            wd = group["weight_decay"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                assert not p.is_cuda, \
                    "DeepSpeedCPUAdam steps host params (use FusedAdam on GPU)"
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    master = p if p.dtype == torch.float32 else p.float()
                    if p.dtype != torch.float32:
                        state["master"] = master.detach().clone()
                    state["exp_avg"] = torch.zeros_like(
                        master, dtype=torch.float32)
                    state["exp_avg_sq"] = torch.zeros_like(
                        master, dtype=torch.float32)
                state["step"] += 1
                master = state.get("master", p)
                g = p.grad.contiguous().view(-1)
                if ext is not None and hasattr(ext, "cpu_adam_flat") and \
                        g.dtype in (torch.float32, torch.bfloat16):
                    w16 = p.view(-1) if p.dtype == torch.bfloat16 else None
                    ext.cpu_adam_flat(master.view(-1), g,
                                      state["exp_avg"].view(-1),
                                      state["exp_avg_sq"].view(-1), w16,
                                      lr, beta1, beta2, eps, wd,
                                      state["step"], 1.0, self.adam_w_mode)
                    if p.dtype not in (torch.bfloat16, torch.float32):
                        p.copy_(master)
                else:
                    _torch_adam_step(master, p.grad, state["exp_avg"],
                                     state["exp_avg_sq"], lr, beta1, beta2,
                                     eps, wd, state["step"],
                                     self.adam_w_mode)
                    if master is not p:
                        p.copy_(master)
        return loss
