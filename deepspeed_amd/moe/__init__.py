"""Mixture-of-Experts with expert parallelism over xGMI all-to-all."""

from .experts import Experts
from .layer import (MoE, has_moe_layers, is_moe_param,
                    split_params_into_different_moe_groups_for_optimizer)
from .sharded_moe import MOELayer, TopKGate, topkgating

__all__ = ["MoE", "Experts", "MOELayer", "TopKGate", "topkgating",
           "has_moe_layers", "is_moe_param",
           "split_params_into_different_moe_groups_for_optimizer"]
