"""Local expert containers (reference deepspeed/moe/experts.py).

``Experts``: each EP rank owns ``num_local_experts`` deep copies of the
expert module, executed as a Python loop — the generic fallback for
arbitrary expert architectures.

``FusedExperts``: the MI355X grouped-GEMM container (role of the
reference's CUTLASS grouped GEMM,
inference/v2/kernels/cutlass_ops/moe_gemm/moe_gemm.cu:175). Expert
weights live as stacked 3D parameters [E_local, out, in] and all local
experts run in ONE strided-batched hipBLASLt GEMM per projection
(torch.bmm) plus the fused gated-activation HIP kernel — 3 launches
instead of 3*E_local. Used automatically when the expert module is a
gate/up/down SwiGLU MLP (LlamaMLP-shaped).

Every expert parameter is tagged ``allreduce=False`` + ``group_name`` so the
ZeRO optimizer reduces it over the expert-data-parallel group instead of
the full DP group.
"""

import copy

import torch
import torch.nn as nn

from ..ops import swiglu


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


def is_swiglu_mlp(m: nn.Module) -> bool:
    """True for gate/up/down bias-free MLPs (LlamaMLP & friends)."""
    return all(
        isinstance(getattr(m, n, None), nn.Linear)
        and getattr(m, n).bias is None for n in
        ("gate_proj", "up_proj", "down_proj"))


class FusedExperts(nn.Module):
    """Grouped expert GEMMs over stacked 3D weights (see module docstring).

    forward consumes the already-dispatched [E_local, N, d_model] batch
    (capacity-packed rows from the index dispatch) and returns the same
    shape — MOELayer feeds it exactly what the per-expert loop would see.
    """

    def __init__(self, template: nn.Module, num_local_experts: int,
                 expert_group_name=None):
        super().__init__()
        d_ff, d_model = template.gate_proj.weight.shape
        self.num_local_experts = num_local_experts
        self.d_model, self.d_ff = d_model, d_ff
        dt = template.gate_proj.weight.dtype
        e = num_local_experts
        # weights stored PRE-TRANSPOSED ([E, in, out]) so every bmm gets a
        # contiguous B operand — besides skipping the strided-B path,
        # transposed-view B at large shapes hit a hipBLASLt memory fault
        # (r2 call 15: [8,5120,1024]x[8,1024,2816] bf16)
        self.w_gate = nn.Parameter(torch.empty(e, d_model, d_ff, dtype=dt))
        self.w_up = nn.Parameter(torch.empty(e, d_model, d_ff, dtype=dt))
        self.w_down = nn.Parameter(torch.empty(e, d_ff, d_model, dtype=dt))
        self.reset_parameters()
        # expert 0 keeps the template's weights (parity with deepcopy init)
        with torch.no_grad():
            self.w_gate[0].copy_(template.gate_proj.weight.t())
            self.w_up[0].copy_(template.up_proj.weight.t())
            self.w_down[0].copy_(template.down_proj.weight.t())
        for p in self.parameters():
            p.allreduce = False
            p.group_name = expert_group_name

    def expert_parameters(self, j):
        """Per-expert views in LlamaMLP parameter order and SHAPE
        ([out, in], transposed views of the stored [in, out] weights) —
        used by tests/tools that address experts individually."""
        return [self.w_gate[j].t(), self.w_up[j].t(), self.w_down[j].t()]

    def reset_parameters(self, std: float = 0.02):
        for w in (self.w_gate, self.w_up, self.w_down):
            w.data.normal_(0.0, std)

    # This torch/hipBLASLt build memory-faults on bf16 bmm at large
    # grouped shapes (probe r2_call16: [8,5120,1024]x[8,1024,2816] faults
    # with BOTH contiguous and strided B; [8,512,1024] works). The grouped
    # launch only matters when per-expert GEMMs are small/launch-bound, so
    # route: bmm below the validated boundary, per-expert full-tile GEMMs
    # (one hipBLASLt launch each, compute-bound anyway) above it.
    _BMM_MAX_TOKENS = 512

    def forward(self, x):
        # x: [E_local, N, d_model]; weights already [E, in, out]
        if x.size(1) <= self._BMM_MAX_TOKENS or not x.is_cuda:
            g = torch.bmm(x, self.w_gate)
            u = torch.bmm(x, self.w_up)
            h = swiglu(g, u)
            return torch.bmm(h, self.w_down)
        outs = []
        for e in range(self.num_local_experts):
            h = swiglu(x[e] @ self.w_gate[e], x[e] @ self.w_up[e])
            outs.append(h @ self.w_down[e])
        return torch.stack(outs, dim=0)
