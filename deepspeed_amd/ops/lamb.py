"""LAMB optimizer (reference: deepspeed/ops/lamb/fused_lamb.py over
csrc/lamb/fused_lamb_cuda_kernel.cu).

Layerwise trust-ratio Adam for very large batch sizes. On GPU fp32 params the
step runs as the hand-written 2-phase HIP kernel (csrc/optim.hip: Adam
direction with fused squared-norm reductions, then trust-ratio apply);
other dtypes/devices fall back to the torch composition below.
"""

import torch

from ._loader import get_ext


class FusedLamb(torch.optim.Optimizer):
    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-6,
                 weight_decay=0.0, bias_correction=True, max_coeff=10.0,
                 min_coeff=0.01):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay,
                        bias_correction=bias_correction,
                        max_coeff=max_coeff, min_coeff=min_coeff)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            beta1, beta2 = group["betas"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                g = p.grad.float()
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(p,
                                                        dtype=torch.float32)
                    state["exp_avg_sq"] = torch.zeros_like(p,
                                                           dtype=torch.float32)
                state["step"] += 1
                m, v = state["exp_avg"], state["exp_avg_sq"]
                ext = get_ext()
                if (ext is not None and p.is_cuda
                        and p.dtype == torch.float32
                        and group["bias_correction"]):
                    # hand-written 2-phase HIP kernel (csrc/optim.hip):
                    # Adam direction + fused norm reductions, then the
                    # trust-ratio apply — no host sync for the norms
                    if "u_buf" not in state:
                        state["u_buf"] = torch.empty_like(p)
                        state["norms2"] = torch.zeros(
                            2, dtype=torch.float32, device=p.device)
                    ext.fused_lamb(p.data, p.grad.contiguous(), m, v,
                                   state["u_buf"], state["norms2"], None,
                                   group["lr"], beta1, beta2, group["eps"],
                                   group["weight_decay"], state["step"],
                                   group["max_coeff"], group["min_coeff"],
                                   1.0)
                    continue
                m.mul_(beta1).add_(g, alpha=1 - beta1)
                v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
                if group["bias_correction"]:
                    bc1 = 1 - beta1 ** state["step"]
                    bc2 = 1 - beta2 ** state["step"]
                else:
                    bc1 = bc2 = 1.0
                update = (m / bc1) / ((v / bc2).sqrt() + group["eps"])
                if group["weight_decay"] != 0.0:
                    update = update + group["weight_decay"] * p.float()
                w_norm = p.detach().float().norm()
                u_norm = update.norm()
                if w_norm > 0 and u_norm > 0:
                    trust = (w_norm / u_norm).clamp(group["min_coeff"],
                                                    group["max_coeff"])
                else:
                    trust = torch.ones((), device=p.device)
                p.add_((-group["lr"] * trust * update).to(p.dtype))
        return loss
