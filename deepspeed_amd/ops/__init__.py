"""Hand-written CDNA4 (gfx950) HIP ops with CPU torch fallbacks."""

from ._loader import has_ext, get_ext
from .adam import FusedAdam, fused_adam_step, multi_tensor_adam_available
from .norms import RMSNorm, FusedLayerNorm, rms_norm, layer_norm
from .rope import apply_rope, rope_tables
from .swiglu import swiglu, geglu
from .evoformer import DS4Sci_EvoformerAttention, EvoformerAttention
from .spatial import nhwc_bias_add
from .token_ops import token_gather, token_scatter

__all__ = [
    "has_ext", "get_ext", "FusedAdam", "fused_adam_step",
    "multi_tensor_adam_available", "RMSNorm", "FusedLayerNorm", "rms_norm",
    "layer_norm", "apply_rope", "rope_tables", "swiglu", "geglu",
    "nhwc_bias_add", "token_gather", "token_scatter",
]
