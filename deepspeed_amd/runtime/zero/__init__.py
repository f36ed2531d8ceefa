"""ZeRO optimizers: stage 1/2 (flat-bucket grad/state partitioning) and
stage 3 (module-unit parameter partitioning) + the Init/GatheredParameters
public API."""

from .partition import GatheredParameters, Init
from .stage12 import ZeroStage12Optimizer
from .stage3 import ZeroStage3Optimizer

__all__ = ["ZeroStage12Optimizer", "ZeroStage3Optimizer", "Init",
           "GatheredParameters"]
