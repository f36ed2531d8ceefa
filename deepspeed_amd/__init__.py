"""deepspeed_amd — MI355X-native distributed training framework.

Brand-new implementation with the capability surface of microsoft/DeepSpeed
(see SURVEY.md), built for AMD Instinct MI355X (gfx950/CDNA4): PyTorch-ROCm
tensors, hand-written HIP kernels for the hot ops, RCCL collectives over the
8-GPU xGMI mesh.

Public API parity: ``initialize()`` (reference deepspeed/__init__.py:69),
``init_inference()`` (:291), ``add_config_arguments()`` (:268), plus the
``comm``, ``zero``, ``ops`` submodules.
"""

__version__ = "0.1.0"

from typing import Optional, Union

import torch

from . import accel
from . import comm
from .config import Config
from .runtime.engine import Engine
from .utils.logging import logger, log_dist

# convenience re-exports
from .ops.adam import FusedAdam  # noqa: F401
from .ops.norms import RMSNorm, FusedLayerNorm  # noqa: F401
from .ops.transformer import (DeepSpeedTransformerConfig,  # noqa: F401
                              DeepSpeedTransformerLayer)


def initialize(args=None,
               model: torch.nn.Module = None,
               optimizer: Optional[torch.optim.Optimizer] = None,
               model_parameters=None,
               training_data=None,
               lr_scheduler=None,
               mpu=None,
               dist_init_required: Optional[bool] = None,
               collate_fn=None,
               config: Union[str, dict, None] = None,
               config_params=None):
    """Initialize the training engine.

    Returns ``(engine, optimizer, training_dataloader, lr_scheduler)`` —
    the reference's 4-tuple contract.
    """
    assert model is not None, "deepspeed_amd.initialize: model is required"
    if config is None and config_params is not None:
        config = config_params
    if config is None and args is not None and \
            getattr(args, "deepspeed_config", None):
        config = args.deepspeed_config

    if dist_init_required is None or dist_init_required:
        if not comm.is_initialized():
            comm.init_distributed()

    cfg = Config(config, world_size=comm.get_world_size())

    # sequence-parallel mesh, if requested in the config
    sp = cfg.raw.get("sequence_parallel_size", 1)
    if sp > 1:
        from .parallel import groups
        groups.initialize_sequence_parallel(sp)

    from .runtime.pipe.module import PipelineModule
    if isinstance(model, PipelineModule):
        from .runtime.pipe.engine import PipelineEngine
        engine = PipelineEngine(model=model,
                                optimizer=optimizer,
                                model_parameters=model_parameters,
                                lr_scheduler=lr_scheduler,
                                config=cfg)
    else:
        engine = Engine(model=model,
                        optimizer=optimizer,
                        model_parameters=model_parameters,
                        lr_scheduler=lr_scheduler,
                        config=cfg,
                        mpu=mpu)

    dataloader = None
    if training_data is not None:
        dataloader = engine.deepspeed_io(training_data, collate_fn=collate_fn)

    return engine, engine.optimizer, dataloader, engine.lr_scheduler


def init_inference(model: torch.nn.Module, config=None, **kwargs):
    """Build an inference engine (TP sharding + fused kernels + KV cache)."""
    from .inference.engine import InferenceEngine, InferenceConfig
    if isinstance(config, dict):
        cfg = InferenceConfig(**{**config, **kwargs})
    elif config is None:
        cfg = InferenceConfig(**kwargs)
    else:
        cfg = config
    return InferenceEngine(model, cfg)


def add_config_arguments(parser):
    """Add --deepspeed / --deepspeed_config args (reference :268)."""
    group = parser.add_argument_group("DeepSpeed-AMD",
                                      "MI355X training configuration")
    group.add_argument("--deepspeed", default=False, action="store_true",
                       help="Enable the deepspeed_amd engine")
    group.add_argument("--deepspeed_config", default=None, type=str,
                       help="Path to the JSON config")
    group.add_argument("--deescale_config", default=None, type=str,
                       help=argparse_suppress())
    group.add_argument("--local_rank", default=-1, type=int,
                       help="Local rank set by the launcher")
    return parser


def argparse_suppress():
    import argparse
    return argparse.SUPPRESS


def zero_init(config=None, **kwargs):
    """ZeRO-3 construction-time context (reference zero.Init)."""
    from .runtime.zero.partition import Init
    return Init(config=config, **kwargs)


from .runtime import zero  # noqa: E402  (deepspeed.zero parity namespace)


def tp_model_init(model, tp_size: int, dtype=None, config=None, **kwargs):
    """Tensor-parallel training init (reference deepspeed/__init__.py:369)."""
    from .runtime.tensor_parallel import tp_model_init as _tp
    return _tp(model, tp_size, dtype)
