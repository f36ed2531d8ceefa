"""1-bit Adam — error-feedback sign-compressed momentum communication
(reference: deepspeed/runtime/fp16/onebit/adam.py OnebitAdam :14,
runtime/comm/compressed.py compressed_allreduce :13).

Algorithm (NeurIPS'21 "1-bit Adam"): run plain Adam for ``freeze_step``
warmup steps; afterwards freeze the variance ``v`` and communicate only the
*momentum* as 1 bit/element + one fp scale per chunk, with local error
feedback so compression noise cancels over steps. Communication volume
drops ~26x vs fp32 allreduce.

MI355X note: the compressed exchange is a single ``all_gather`` of packed
sign bytes (+ scales) over the DP group — an all-to-all-shaped pattern the
full xGMI mesh serves at line rate; there is no tree/hierarchy to tune on
one node.
"""

import torch

from ... import comm as dist

_POW2 = torch.tensor([1, 2, 4, 8, 16, 32, 64, 128], dtype=torch.uint8)


def pack_signs(x: torch.Tensor) -> torch.Tensor:
    """sign(x) -> packed uint8 (1 bit/elem, little-endian within byte).
    numel must be a multiple of 8."""
    bits = (x >= 0).to(torch.uint8).reshape(-1, 8)
    return (bits * _POW2.to(x.device)).sum(dim=1, dtype=torch.uint8)


def unpack_signs(packed: torch.Tensor, n: int) -> torch.Tensor:
    """packed uint8 -> {-1, +1} float tensor of length n."""
    b = packed.reshape(-1, 1).bitwise_and(_POW2.to(packed.device)).ne(0)
    return b.reshape(-1)[:n].float().mul_(2.0).sub_(1.0)


def compressed_allreduce(x: torch.Tensor, error: torch.Tensor, group=None):
    """Average ``x`` across the group at 1 bit/element with error feedback.

    Returns the averaged tensor (same shape); ``error`` is updated in place
    with the local compression residual (reference compressed_allreduce:51).
    """
    world = dist.get_world_size(group)
    n = x.numel()
    flat = x.reshape(-1) + error.reshape(-1)
    pad = (-n) % 8
    if pad:
        flat = torch.cat([flat, flat.new_zeros(pad)])
    scale = flat.abs().mean()
    comp = pack_signs(flat)
    # local error feedback: what the wire loses stays here for next step
    decomp_local = unpack_signs(comp, n) * scale
    error.reshape(-1).copy_(x.reshape(-1) + error.reshape(-1) - decomp_local)

    if world == 1:
        return decomp_local.reshape(x.shape)

    gathered = [torch.empty_like(comp) for _ in range(world)]
    scales = torch.empty(world, dtype=scale.dtype, device=x.device)
    dist.all_gather(gathered, comp, group=group)
    dist.all_gather_into_tensor(scales, scale.reshape(1), group=group)
    out = torch.zeros(n, dtype=torch.float32, device=x.device)
    for r in range(world):
        out += unpack_signs(gathered[r], n) * scales[r]
    out /= world
    return out.reshape(x.shape)


class OnebitAdam(torch.optim.Optimizer):
    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0.0, freeze_step=100, deepspeed=None,
                 comm_group=None):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)
        self.freeze_step = freeze_step
        self.comm_group = comm_group
        self.adam_freeze_key = False  # True once in the compressed stage

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        world = dist.get_world_size(self.comm_group) \
            if dist.is_initialized() else 1
        for group in self.param_groups:
            beta1, beta2 = group["betas"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                g = p.grad.float()
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(p, dtype=torch.float32)
                    state["exp_avg_sq"] = torch.zeros_like(p,
                                                           dtype=torch.float32)
                    state["error"] = torch.zeros_like(p, dtype=torch.float32)
                state["step"] += 1
                m, v = state["exp_avg"], state["exp_avg_sq"]

                if state["step"] <= self.freeze_step:
                    # warmup: exact Adam; grads averaged by the caller (DDP/
                    # ZeRO) or here if used standalone
                    if world > 1:
                        dist.all_reduce(g, group=self.comm_group)
                        g /= world
                    m.mul_(beta1).add_(g, alpha=1 - beta1)
                    v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
                    if state["step"] == self.freeze_step:
                        self.adam_freeze_key = True
                else:
                    # compressed stage: v frozen, momentum averaged at
                    # 1 bit/elem with error feedback
                    m.mul_(beta1).add_(g, alpha=1 - beta1)
                    m.copy_(compressed_allreduce(m, state["error"],
                                                 self.comm_group))

                bc1 = 1 - beta1 ** state["step"]
                bc2 = 1 - beta2 ** state["step"]
                denom = (v / bc2).sqrt_().add_(group["eps"])
                update = (m / bc1) / denom
                if group["weight_decay"] != 0.0:
                    update = update.add(p.float(), alpha=group["weight_decay"])
                p.add_(update.to(p.dtype), alpha=-group["lr"])
        return loss


class ZeroOneAdam(torch.optim.Optimizer):
    """0/1 Adam (reference fp16/onebit/zoadam.py): both the variance AND
    the communication are intermittent — variance refreshes every
    ``var_update_interval`` steps and momenta synchronize (1-bit
    compressed) every ``local_step_interval`` steps, with error feedback
    carrying the skipped information. Cuts communication ROUNDS as well as
    volume vs 1-bit Adam."""

    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0.0, var_freeze_step=50, var_update_scaler=4,
                 local_step_scaler=2, comm_group=None):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)
        self.var_freeze_step = var_freeze_step
        self.var_update_scaler = var_update_scaler
        self.local_step_scaler = local_step_scaler
        self.comm_group = comm_group

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        world = dist.get_world_size(self.comm_group) \
            if dist.is_initialized() else 1
        for group in self.param_groups:
            beta1, beta2 = group["betas"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                g = p.grad.float()
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(p, dtype=torch.float32)
                    state["exp_avg_sq"] = torch.zeros_like(p,
                                                           dtype=torch.float32)
                    state["error"] = torch.zeros_like(p, dtype=torch.float32)
                state["step"] += 1
                t = state["step"]
                m, v = state["exp_avg"], state["exp_avg_sq"]

                frozen = t > self.var_freeze_step
                if not frozen:
                    # exact stage: synchronous grads, live variance
                    if world > 1:
                        dist.all_reduce(g, group=self.comm_group)
                        g /= world
                    m.mul_(beta1).add_(g, alpha=1 - beta1)
                    v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
                else:
                    m.mul_(beta1).add_(g, alpha=1 - beta1)
                    # intermittent variance refresh (local, frozen between)
                    if t % self.var_update_scaler == 0:
                        v.mul_(beta2).addcmul_(m, m, value=1 - beta2)
                    # intermittent compressed momentum sync
                    if t % self.local_step_scaler == 0:
                        m.copy_(compressed_allreduce(m, state["error"],
                                                     self.comm_group))

                bc1 = 1 - beta1 ** t
                bc2 = 1 - beta2 ** t
                denom = (v / bc2).sqrt_().add_(group["eps"])
                update = (m / bc1) / denom
                if group["weight_decay"] != 0.0:
                    update = update.add(p.float(), alpha=group["weight_decay"])
                p.add_(update.to(p.dtype), alpha=-group["lr"])
        return loss


class OnebitLamb(torch.optim.Optimizer):
    """1-bit LAMB (reference deepspeed/runtime/fp16/onebit/lamb.py).

    Warmup stage: exact LAMB with dense gradient all-reduce. Compressed
    stage: the variance term and per-tensor scaling coefficients freeze
    at their warmup values, and only the momentum is exchanged — sign +
    per-tensor scale with error feedback (26x less traffic than fp32
    all-reduce), the same compressed_allreduce as OnebitAdam.
    """

    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-6,
                 weight_decay=0.0, freeze_step=100, max_coeff=10.0,
                 min_coeff=0.01, deepspeed=None, comm_group=None):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay, max_coeff=max_coeff,
                        min_coeff=min_coeff)
        super().__init__(params, defaults)
        self.freeze_step = freeze_step
        self.comm_group = comm_group
        self.lamb_freeze_key = False

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        world = dist.get_world_size(self.comm_group) \
            if dist.is_initialized() else 1
        for group in self.param_groups:
            beta1, beta2 = group["betas"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                g = p.grad.float()
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(p, dtype=torch.float32)
                    state["exp_avg_sq"] = torch.zeros_like(p,
                                                           dtype=torch.float32)
                    state["error"] = torch.zeros_like(p, dtype=torch.float32)
                    state["frozen_trust"] = None
                state["step"] += 1
                m, v = state["exp_avg"], state["exp_avg_sq"]

                if state["step"] <= self.freeze_step:
                    if world > 1:
                        dist.all_reduce(g, group=self.comm_group)
                        g /= world
                    m.mul_(beta1).add_(g, alpha=1 - beta1)
                    v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
                    if state["step"] == self.freeze_step:
                        self.lamb_freeze_key = True
                else:
                    # compressed stage: v (and the trust ratio) frozen
                    m.mul_(beta1).add_(g, alpha=1 - beta1)
                    m.copy_(compressed_allreduce(m, state["error"],
                                                 self.comm_group))

                bc1 = 1 - beta1 ** state["step"]
                bc2 = 1 - beta2 ** state["step"]
                update = (m / bc1) / ((v / bc2).sqrt().add_(group["eps"]))
                if group["weight_decay"] != 0.0:
                    update = update.add(p.float(), alpha=group["weight_decay"])
                if state["step"] <= self.freeze_step:
                    w_norm = p.detach().float().norm()
                    u_norm = update.norm()
                    if w_norm > 0 and u_norm > 0:
                        trust = (w_norm / u_norm).clamp(group["min_coeff"],
                                                        group["max_coeff"])
                    else:
                        trust = torch.ones((), device=p.device)
                    state["frozen_trust"] = trust
                else:
                    trust = state["frozen_trust"] if \
                        state["frozen_trust"] is not None else 1.0
                p.add_((-group["lr"] * trust * update).to(p.dtype))
        return loss
