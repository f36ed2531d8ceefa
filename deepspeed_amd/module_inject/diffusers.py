"""Diffusion (stable-diffusion UNet) transformer-block injection
(reference: deepspeed/ops/transformer/inference/
diffusers_transformer_block.py DeepSpeedDiffusersTransformerBlock, wired
by module_inject/replace_module.py for diffusers pipelines).

Replaces a diffusers ``BasicTransformerBlock``-shaped module (duck-typed:
norm1/2/3 LayerNorms, attn1/attn2, ff.net = [GEGLU(proj), dropout,
Linear]) with a block that runs this framework's fused kernels: HIP
LayerNorm, fused GEGLU, and the channels-last fused bias+residual add
(ops/csrc/spatial.hip) for the epilogues. Attention modules are kept and
route through whatever attention path they already use (the sdpa swap in
replace.py applies to them independently)."""

import torch
import torch.nn as nn

from ..ops.norms import layer_norm
from ..ops.swiglu import geglu
from ..ops.spatial import nhwc_bias_add


class DiffusersTransformerBlock(nn.Module):
    def __init__(self, block: nn.Module):
        super().__init__()
        for name in ("norm1", "norm2", "norm3"):
            src = getattr(block, name)
            setattr(self, f"{name}_g", nn.Parameter(src.weight.data,
                                                    requires_grad=False))
            setattr(self, f"{name}_b", nn.Parameter(src.bias.data,
                                                    requires_grad=False))
            setattr(self, f"{name}_eps", src.eps)
        self.attn1 = block.attn1
        self.attn2 = block.attn2
        proj = block.ff.net[0].proj
        out = block.ff.net[2]
        self.ff1_w = nn.Parameter(proj.weight.data, requires_grad=False)
        self.ff1_b = nn.Parameter(proj.bias.data, requires_grad=False)
        self.ff2_w = nn.Parameter(out.weight.data, requires_grad=False)
        self.ff2_b = nn.Parameter(out.bias.data, requires_grad=False)

    def forward(self, hidden_states, context=None, timestep=None, **kwargs):
        # diffusers >= 0.11 passes encoder_hidden_states instead of context
        if kwargs.get("encoder_hidden_states") is not None:
            context = kwargs["encoder_hidden_states"]
        h = layer_norm(hidden_states, self.norm1_g, self.norm1_b,
                       self.norm1_eps)
        a1 = self.attn1(h) + hidden_states
        h = layer_norm(a1, self.norm2_g, self.norm2_b, self.norm2_eps)
        a2 = self.attn2(h, context) + a1
        h = layer_norm(a2, self.norm3_g, self.norm3_b, self.norm3_eps)
        ff = nn.functional.linear(h, self.ff1_w) + self.ff1_b
        # diffusers GEGLU: value half first, gate half second
        up, gate = ff.chunk(2, dim=-1)
        ff = nn.functional.linear(geglu(gate.contiguous(), up.contiguous()),
                                  self.ff2_w)
        return nhwc_bias_add(ff, self.ff2_b, other=a2)


def _looks_like_basic_transformer_block(m: nn.Module) -> bool:
    try:
        return (isinstance(m.norm1, nn.LayerNorm)
                and isinstance(m.norm2, nn.LayerNorm)
                and isinstance(m.norm3, nn.LayerNorm)
                and m.attn1 is not None and m.attn2 is not None
                and hasattr(m.ff.net[0], "proj")
                and isinstance(m.ff.net[2], nn.Linear))
    except (AttributeError, IndexError, TypeError):
        return False


def replace_diffusers_blocks(model: nn.Module) -> int:
    """Swap every BasicTransformerBlock-shaped submodule for the fused
    block. Returns the number of replacements."""
    n = 0
    for parent in model.modules():
        for name, child in list(parent.named_children()):
            if _looks_like_basic_transformer_block(child):
                setattr(parent, name, DiffusersTransformerBlock(child))
                n += 1
    return n
