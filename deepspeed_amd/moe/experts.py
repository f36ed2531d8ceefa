"""Local expert container (reference deepspeed/moe/experts.py).

Each EP rank owns ``num_local_experts`` deep copies of the expert module.
Every expert parameter is tagged ``allreduce=False`` + ``group_name`` so the
ZeRO optimizer reduces it over the expert-data-parallel group instead of
the full DP group.
"""

import copy

import torch.nn as nn


class Experts(nn.Module):
    def __init__(self, expert: nn.Module, num_local_experts: int = 1,
                 expert_group_name=None):
        super().__init__()
        self.local_experts = nn.ModuleList(
            [copy.deepcopy(expert) for _ in range(num_local_experts)])
        self.num_local_experts = num_local_experts
        for e in self.local_experts:
            for p in e.parameters():
                p.allreduce = False
                p.group_name = expert_group_name

    def forward(self, x):
        # used only for the single-expert fast path; MOELayer batches itself
        return self.local_experts[0](x)
