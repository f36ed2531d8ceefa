"""User-facing MoE wrapper (reference deepspeed/moe/layer.py MoE :17)."""

from typing import Optional

import torch.nn as nn

from .. import comm as dist
from ..parallel import groups
from ..utils.logging import log_dist
from .experts import Experts, FusedExperts, is_swiglu_mlp
from .sharded_moe import MOELayer, TopKGate


class MoE(nn.Module):
    """Wraps an expert module into an expert-parallel MoE layer.

    Args mirror the reference: hidden_size, expert (module to replicate),
    num_experts, ep_size, k, capacity_factor, eval_capacity_factor,
    min_capacity, noisy_gate_policy, drop_tokens, use_residual.
    """

    def __init__(self,
                 hidden_size: int,
                 expert: nn.Module,
                 num_experts: int = 1,
                 ep_size: int = 1,
                 k: int = 1,
                 capacity_factor: float = 1.0,
                 eval_capacity_factor: float = 1.0,
                 min_capacity: int = 4,
                 use_residual: bool = False,
                 noisy_gate_policy: Optional[str] = None,
                 drop_tokens: bool = True,
                 use_rts: bool = True,
                 use_tutel: bool = False,
                 enable_expert_tensor_parallelism: bool = False,
                 top2_2nd_expert_sampling: bool = True,
                 use_fused_experts: bool = True):
        super().__init__()
        self.use_residual = use_residual
        self.ep_size = ep_size
        self.num_experts = num_experts
        assert num_experts % ep_size == 0, \
            f"num_experts ({num_experts}) not divisible by ep_size ({ep_size})"
        self.num_local_experts = num_experts // ep_size
        self.expert_group_name = f"ep_size_{ep_size}"

        # gate first: FusedExperts' init draws from the global RNG, and the
        # gate weights must not depend on the expert container choice
        gate = TopKGate(hidden_size, num_experts, k, capacity_factor,
                        eval_capacity_factor, min_capacity, noisy_gate_policy,
                        drop_tokens, use_rts, None, top2_2nd_expert_sampling)
        if use_fused_experts and is_swiglu_mlp(expert):
            # grouped-GEMM container: one bmm per projection for all
            # local experts (MI355X stand-in for CUTLASS moe_gemm)
            experts = FusedExperts(expert, self.num_local_experts,
                                   self.expert_group_name)
        else:
            experts = Experts(expert, self.num_local_experts,
                              self.expert_group_name)
        self.deepspeed_moe = MOELayer(gate, experts, self.expert_group_name,
                                      self.ep_size, self.num_local_experts)
        if use_residual:
            import copy
            self.mlp = copy.deepcopy(expert)
            self.coefficient = nn.Linear(hidden_size, 2)
        self._ep_initialized = False
        if dist.is_initialized():
            self._create_process_groups()

    def _create_process_groups(self):
        if self._ep_initialized:
            return
        world = dist.get_world_size()
        ep = min(self.ep_size, world)
        if ep != self.ep_size:
            # clamp like the reference when world < ep_size
            assert self.num_experts % ep == 0
            self.ep_size = ep
            self.deepspeed_moe.ep_size = ep
            self.expert_group_name = f"ep_size_{ep}"
            self.deepspeed_moe.ep_group_name = self.expert_group_name
        groups.initialize_expert_parallel(self.ep_size, self.expert_group_name)
        self.deepspeed_moe._set_ep_group(
            groups.get_expert_parallel_group(self.expert_group_name))
        for p in self.deepspeed_moe.experts.parameters():
            p.group_name = self.expert_group_name
        self._ep_initialized = True
        log_dist(f"MoE layer: experts={self.num_experts} ep={self.ep_size} "
                 f"local_experts={self.num_local_experts}")

    def set_deepspeed_parallelism(self, use_data_before_expert_parallel_=False):
        self._create_process_groups()

    def forward(self, hidden_states, used_token=None):
        if not self._ep_initialized and dist.is_initialized():
            self._create_process_groups()
        output = self.deepspeed_moe(hidden_states, used_token)
        if self.use_residual:
            res = self.mlp(hidden_states)
            if isinstance(res, tuple):
                res = res[0]
            coef = self.coefficient(hidden_states).softmax(dim=-1)
            output = output * coef[..., 0:1] + res * coef[..., 1:]
        return output, self.deepspeed_moe.l_aux, self.deepspeed_moe.exp_counts


def has_moe_layers(module) -> bool:
    for m in module.modules():
        if isinstance(m, MoE):
            return True
    return False


def is_moe_param(p) -> bool:
    return getattr(p, "allreduce", True) is False


def split_params_into_different_moe_groups_for_optimizer(param_groups):
    """Split optimizer param groups so every expert group gets its own
    entry tagged with ``moe=True`` + ``name`` (reference
    deepspeed/moe/utils.py:69)."""
    if isinstance(param_groups, dict):
        param_groups = [param_groups]
    out = []
    for group in param_groups:
        dense = [p for p in group["params"] if not is_moe_param(p)]
        moe_by_name = {}
        for p in group["params"]:
            if is_moe_param(p):
                moe_by_name.setdefault(p.group_name, []).append(p)
        g0 = dict(group)
        g0["params"] = dense
        out.append(g0)
        for name, ps in moe_by_name.items():
            g = dict(group)
            g["params"] = ps
            g["moe"] = True
            g["name"] = name
            out.append(g)
    return out
