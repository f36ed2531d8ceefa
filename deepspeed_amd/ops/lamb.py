"""LAMB optimizer (reference: deepspeed/ops/lamb/fused_lamb.py over
csrc/lamb/fused_lamb_cuda_kernel.cu).

Layerwise trust-ratio Adam for very large batch sizes. The per-parameter
norm reductions + elementwise update run as a handful of torch kernels on
ROCm; the flat-shard fused-kernel treatment (like adam.hip) is only worth
it if LAMB becomes a bench path.
"""

import torch


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
