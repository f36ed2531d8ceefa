"""Expert-parallel MoE layer: top-k gating, all-to-all dispatch, local experts.

Capability parity with the reference's ``deepspeed/moe/sharded_moe.py``
(MOELayer :533, top1gating :183, top2gating :290, topkgating :374,
_AllToAll :96) — re-designed for the MI355X node:

* **Dispatch by index, not one-hot einsum.** The reference builds
  [tokens, experts, capacity] one-hot dispatch masks and einsums against
  them (O(T·E·C) memory traffic). Here routing produces a flat destination
  index per (token, k) and dispatch/combine are ``index_add_`` / ``gather``
  — pure HBM-bandwidth ops with no dead zeros, which is what an
  8 TB/s-bound chip wants.
* **One all_to_all_single per direction** on a contiguous
  [experts·capacity, d_model] buffer over the EP group. The 8-GPU xGMI
  mesh is fully connected (7 p2p links/GPU), so a2a uses every link
  simultaneously — it is the best-mapped collective on this topology and
  the reason EP degree 8 is the default for MoE on one node.
* Gating runs in fp32 (tiny GEMM); aux load-balancing loss follows GShard
  (l_aux = E · Σ_e me_e · ce_e).
"""

from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import comm as dist

uniform_map = {}
gumbel_map = {}
exp_selection_uniform_map = {}


class _AllToAll(torch.autograd.Function):
    """Differentiable all_to_all_single (reference sharded_moe.py:96)."""

    @staticmethod
    def forward(ctx, group, input_):
        ctx.group = group
        input_ = input_.contiguous()
        if group is None or dist.get_world_size(group) == 1:
            return input_
        output = torch.empty_like(input_)
        dist.all_to_all_single(output, input_, group=group)
        return output

    @staticmethod
    def backward(ctx, grad_output):
        return None, _AllToAll.apply(ctx.group, grad_output)


def multiplicative_jitter(x, device, epsilon=1e-2):
    """Uniform multiplicative jitter on gate inputs (reference :142)."""
    if epsilon == 0:
        return x
    uniform = uniform_map.get(device)
    if uniform is None:
        uniform = torch.distributions.uniform.Uniform(
            low=torch.tensor(1.0 - epsilon, device=device),
            high=torch.tensor(1.0 + epsilon, device=device)).rsample
        uniform_map[device] = uniform
    return x * uniform(x.shape)


def gumbel_rsample(shape, device):
    gumbel = gumbel_map.get(device)
    if gumbel is None:
        one = torch.tensor(1.0, device=device)
        zero = torch.tensor(0.0, device=device)
        gumbel = torch.distributions.gumbel.Gumbel(zero, one).rsample
        gumbel_map[device] = gumbel
    return gumbel(shape)


@torch.jit.ignore
def _capacity(num_tokens, num_experts, capacity_factor, min_capacity, k=1):
    cap = int(k * num_tokens / num_experts * capacity_factor)
    return max(cap, min_capacity)


def topkgating(logits: torch.Tensor,
               k: int,
               capacity_factor: float = 1.0,
               min_capacity: int = 4,
               drop_tokens: bool = True,
               ep_group=None,
               noisy_gate_policy: Optional[str] = None,
               training: bool = True):
    """Top-k gating (covers the reference's top1/top2/topk variants).

    Returns (l_aux, combine_info) where combine_info carries the flat
    destination index, kept mask and normalized gate value per (token, k).
    """
    num_tokens, num_experts = logits.shape

    route_logits = logits
    if noisy_gate_policy == "RSample" and training:
        route_logits = logits + gumbel_rsample(logits.shape, logits.device)
    gates = F.softmax(logits, dim=1)

    topk_vals, topk_idx = torch.topk(
        F.softmax(route_logits, dim=1) if noisy_gate_policy else gates,
        k, dim=1)                                   # [T, k]

    capacity = _capacity(num_tokens, num_experts, capacity_factor,
                         min_capacity, k)
    if not drop_tokens:
        # capacity = global max tokens routed to any expert
        counts = torch.bincount(topk_idx.reshape(-1),
                                minlength=num_experts)
        cap_t = counts.max()
        if ep_group is not None and dist.get_world_size(ep_group) > 1:
            dist.all_reduce(cap_t, op=dist.ReduceOp.MAX, group=ep_group)
        capacity = max(int(cap_t.item()), min_capacity)

    # position of each (token,k) inside its expert's capacity buffer:
    # process choices column-major (all k=0 first) so primary routes win
    flat_idx = topk_idx.t().reshape(-1)                       # [k*T]
    one_hot = F.one_hot(flat_idx, num_experts)                # [k*T, E] int
    locations = torch.cumsum(one_hot, dim=0) - 1              # loc within expert
    loc = locations.gather(1, flat_idx.unsqueeze(1)).squeeze(1)
    kept = loc < capacity                                     # drop overflow

    # aux loss: me = mean gate prob, ce = fraction of tokens whose PRIMARY
    # route is expert e (GShard / reference top1gating:258)
    me = gates.mean(dim=0)
    primary = F.one_hot(topk_idx[:, 0], num_experts).float().mean(dim=0)
    l_aux = torch.sum(me * primary) * num_experts

    # gate values for kept routes, renormalized over kept top-k per token
    gate_vals = topk_vals.t().reshape(-1)                     # [k*T] fp32
    gate_vals = gate_vals * kept.to(gate_vals.dtype)
    denom = gate_vals.view(k, num_tokens).sum(dim=0).clamp(min=torch.finfo(gate_vals.dtype).eps)
    gate_vals = gate_vals / denom.repeat(k)

    token_idx = torch.arange(num_tokens, device=logits.device).repeat(k)
    dest = flat_idx * capacity + loc                          # [k*T]
    return l_aux, {
        "capacity": capacity,
        "num_experts": num_experts,
        "dest": dest[kept],
        "token": token_idx[kept],
        "gate": gate_vals[kept],
        "exp_counts": torch.bincount(topk_idx[:, 0], minlength=num_experts).cpu(),
    }


class TopKGate(nn.Module):
    """Gate network (reference TopKGate :480)."""

    def __init__(self, model_dim, num_experts, k=1, capacity_factor=1.0,
                 eval_capacity_factor=1.0, min_capacity=4,
                 noisy_gate_policy: Optional[str] = None, drop_tokens=True,
                 use_rts=True, ep_group=None, top2_2nd_expert_sampling=True):
        super().__init__()
        self.wg = nn.Linear(model_dim, num_experts, bias=False)
        self.ep_group = ep_group
        self.k = k
        self.capacity_factor = capacity_factor
        self.eval_capacity_factor = eval_capacity_factor
        self.min_capacity = min_capacity
        self.noisy_gate_policy = noisy_gate_policy
        self.drop_tokens = drop_tokens
        self.gate_time = 0.0

    def _set_ep_group(self, ep_group):
        self.ep_group = ep_group

    def forward(self, x):
        inp = x.float()
        if self.noisy_gate_policy == "Jitter" and self.training:
            inp = multiplicative_jitter(inp, x.device)
        logits = F.linear(inp, self.wg.weight.float())
        return topkgating(
            logits, self.k,
            self.capacity_factor if self.training else self.eval_capacity_factor,
            self.min_capacity, self.drop_tokens, self.ep_group,
            self.noisy_gate_policy, self.training)


class MOELayer(nn.Module):
    """Mixture-of-experts layer with expert-parallel all-to-all
    (reference MOELayer :533)."""

    def __init__(self, gate: TopKGate, experts: nn.Module, ep_group_name: str,
                 ep_size: int, num_local_experts: int,
                 use_tutel: bool = False):
        super().__init__()
        self.gate = gate
        self.experts = experts
        self.ep_group = None
        self.ep_size = ep_size
        self.ep_group_name = ep_group_name
        self.num_local_experts = num_local_experts
        self.num_experts = ep_size * num_local_experts
        self.l_aux = torch.tensor(0.0)
        self.exp_counts = None

    def _set_ep_group(self, ep_group):
        self.ep_group = ep_group
        self.gate._set_ep_group(ep_group)

    def forward(self, x, *unused):
        d_model = x.shape[-1]
        orig_shape = x.shape
        tokens = x.reshape(-1, d_model)
        T = tokens.shape[0]

        self.l_aux, route = self.gate(tokens)
        self.exp_counts = route["exp_counts"]
        C = route["capacity"]
        E = self.num_experts

        # ---- dispatch: scatter kept tokens into [E*C, M] (index_add is
        # differentiable w.r.t. the source)
        dispatched = tokens.new_zeros((E * C, d_model))
        dispatched.index_add_(0, route["dest"], tokens[route["token"]])

        # ---- all-to-all over EP group: rows [e*C:(e+1)*C] go to e's owner
        dispatched = _AllToAll.apply(self.ep_group, dispatched)

        # ---- local experts: [ep, local_E, C, M] -> per-expert batch
        dispatched = dispatched.reshape(self.ep_size, self.num_local_experts,
                                        C, d_model)
        chunks = dispatched.transpose(0, 1).reshape(
            self.num_local_experts, self.ep_size * C, d_model)
        from .experts import FusedExperts
        if isinstance(self.experts, FusedExperts):
            expert_out = self.experts(chunks)              # [local_E, ep*C, M]
        else:
            outs = []
            for i, expert in enumerate(self.experts.local_experts):
                outs.append(expert(chunks[i]))
            expert_out = torch.stack(outs, dim=0)          # [local_E, ep*C, M]
        expert_out = expert_out.reshape(self.num_local_experts, self.ep_size,
                                        C, d_model).transpose(0, 1)

        # ---- return tokens to their source rank
        expert_out = _AllToAll.apply(self.ep_group,
                                     expert_out.reshape(E * C, d_model))

        # ---- combine: out[token] += gate * expert_out[dest]
        gathered = expert_out[route["dest"]]
        weighted = gathered * route["gate"].unsqueeze(1).to(gathered.dtype)
        out = tokens.new_zeros((T, d_model))
        out.index_add_(0, route["token"], weighted)
        return out.reshape(orig_shape)
