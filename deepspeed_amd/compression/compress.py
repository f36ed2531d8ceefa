"""Compression orchestration (reference: deepspeed/compression/compress.py
init_compression / redundancy_clean, config keyed by module-name groups)."""

import re

import torch.nn as nn

from .basic_layer import LinearLayer_Compress


def _matches(name, patterns):
    return any(re.search(p, name) for p in patterns)


def init_compression(model: nn.Module, compression_config: dict):
    """Replace matching nn.Linear with LinearLayer_Compress and arm the
    configured techniques.

    Config shape (subset of the reference's):
    {"weight_quantization": {"shared_parameters": {...}, "different_groups":
        {"wq1": {"params": {"target_bits": 8, "quantization_group": 64},
                 "modules": ["attention", ...]}}},
     "sparse_pruning": {... "params": {"dense_ratio": 0.7} ...},
     "row_pruning": {...}, "activation_quantization": {...}}
    """
    cc = compression_config or {}
    # accept a full ds_config (reference init_compression takes the whole
    # config and reads its compression_training section)
    if "compression_training" in cc:
        cc = cc["compression_training"] or {}

    def groups_of(section):
        return (cc.get(section, {}) or {}).get("different_groups", {}) or {}

    # collect every module pattern that needs a compress layer
    all_patterns = set()
    for section in ("weight_quantization", "sparse_pruning", "row_pruning",
                    "activation_quantization"):
        for g in groups_of(section).values():
            all_patterns.update(g.get("modules", ["*"]))

    def wants(name):
        return any(p == "*" or re.search(p, name) for p in all_patterns)

    replaced = {}
    for parent_name, parent in list(model.named_modules()):
        for child_name, child in list(parent._modules.items()):
            full = f"{parent_name}.{child_name}" if parent_name else child_name
            if isinstance(child, nn.Linear) and \
                    not isinstance(child, LinearLayer_Compress) and wants(full):
                new = LinearLayer_Compress(child.in_features,
                                           child.out_features,
                                           bias=child.bias is not None)
                new = new.to(child.weight.dtype)
                new.weight.data.copy_(child.weight.data)
                if child.bias is not None:
                    new.bias.data.copy_(child.bias.data)
                parent._modules[child_name] = new
                replaced[full] = new

    def arm(section, fn):
        for g in groups_of(section).values():
            mods = g.get("modules", ["*"])
            params = g.get("params", {})
            for full, layer in replaced.items():
                if any(p == "*" or re.search(p, full) for p in mods):
                    fn(layer, params)

    arm("weight_quantization",
        lambda l, p: l.enable_weight_quantization(
            p.get("target_bits", 8), p.get("quantization_group", 0)))
    arm("activation_quantization",
        lambda l, p: l.enable_activation_quantization(p.get("bits", 8)))
    arm("sparse_pruning",
        lambda l, p: l.enable_sparse_pruning(1.0 - p.get("dense_ratio", 1.0)))
    arm("row_pruning",
        lambda l, p: l.enable_row_pruning(1.0 - p.get("dense_ratio", 1.0)))
    return model


def redundancy_clean(model: nn.Module, compression_config: dict = None):
    """Bake masks/quantization into the weights (reference compress.py)."""
    for m in model.modules():
        if isinstance(m, LinearLayer_Compress):
            m.fix_sparsity()
            m.fix_weight_quantization()
    return model
