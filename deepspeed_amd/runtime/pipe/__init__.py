"""Pipeline parallelism: PipelineModule + 1F1B PipelineEngine over xGMI p2p."""

from .module import LayerSpec, PipelineModule, TiedLayerSpec
from .topology import PipelineParallelGrid

__all__ = ["PipelineModule", "LayerSpec", "TiedLayerSpec",
           "PipelineParallelGrid"]
