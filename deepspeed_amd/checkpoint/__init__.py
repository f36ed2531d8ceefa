from .universal import ds_to_universal

__all__ = ["ds_to_universal"]
