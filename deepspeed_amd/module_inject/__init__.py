from .replace import replace_transformer_layer, HFInjectionPolicy

__all__ = ["replace_transformer_layer", "HFInjectionPolicy"]
