"""Reference-path shim: ``from deepspeed_amd.pipe import PipelineModule``
(reference deepspeed/pipe/__init__.py)."""

from .runtime.pipe.module import (PipelineModule, LayerSpec,  # noqa: F401
                                  TiedLayerSpec)
from .runtime.pipe.topology import PipelineParallelGrid  # noqa: F401
