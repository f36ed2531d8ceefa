from .transformer import DominoTransformerLayer

__all__ = ["DominoTransformerLayer"]
