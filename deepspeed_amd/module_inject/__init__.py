from .replace import replace_transformer_layer, HFInjectionPolicy
from .diffusers import replace_diffusers_blocks, DiffusersTransformerBlock

__all__ = ["replace_transformer_layer", "HFInjectionPolicy",
           "replace_diffusers_blocks", "DiffusersTransformerBlock"]
