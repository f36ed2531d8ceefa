"""Lion optimizer (sign-momentum).

Capability parity with the reference's FusedLion/CPULion
(csrc/lion/multi_tensor_lion.cu, csrc/lion/cpu_lion_impl.cpp). Torch
implementation; the flat ZeRO shards make this memory-bound and simple —
a dedicated HIP kernel is a later optimization.
"""

import torch

from ._loader import get_ext


class Lion(torch.optim.Optimizer):
    """p -= lr * (sign(b1*m + (1-b1)*g) + wd*p);  m = b2*m + (1-b2)*g"""

    def __init__(self, params, lr=1e-4, betas=(0.9, 0.99), weight_decay=0.0):
        defaults = dict(lr=lr, betas=betas, weight_decay=weight_decay)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            lr = group["lr"]
            beta1, beta2 = group["betas"]
            wd = group["weight_decay"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                g = p.grad.float()
                state = self.state[p]
                if len(state) == 0:
                    state["exp_avg"] = torch.zeros_like(p, dtype=torch.float32)
                m = state["exp_avg"]
                ext = get_ext()
                if (ext is not None and p.is_cuda
                        and p.dtype == torch.float32):
                    # fused HIP kernel (csrc/optim.hip); update math is
                    # identical to the torch path below
                    ext.fused_lion(p.data, p.grad.contiguous(), m, None,
                                   lr, beta1, beta2, wd, 1.0)
                    continue
                if wd != 0:
                    p.mul_(1 - lr * wd)
                update = m.mul(beta1).add_(g, alpha=1 - beta1).sign_()
                p.add_(update.to(p.dtype), alpha=-lr)
                m.mul_(beta2).add_(g, alpha=1 - beta2)
        return loss
