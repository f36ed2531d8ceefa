from .optimized_linear import LoRAConfig, OptimizedLinear, QuantizationConfig

__all__ = ["OptimizedLinear", "LoRAConfig", "QuantizationConfig"]
