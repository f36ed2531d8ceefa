from .autotuner import Autotuner

__all__ = ["Autotuner"]
