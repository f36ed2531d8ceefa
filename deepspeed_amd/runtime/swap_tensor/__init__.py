from .swapper import AsyncTensorSwapper

__all__ = ["AsyncTensorSwapper"]
