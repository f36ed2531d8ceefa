"""Typed JSON config system.

Capability parity with the reference's ``deepspeed/runtime/config.py``
(DeepSpeedConfig: JSON/dict -> typed sub-configs with train-batch
reconciliation). Field names keep the reference's JSON schema so existing
DeepSpeed config files work unchanged; defaults are re-tuned for one
MI355X node (8 GPUs, 7 xGMI links each, 288 GB HBM3E per GPU).
"""

import json
from typing import Any, Dict, Optional, Union

from pydantic import BaseModel, Field

from .utils.logging import logger

# xGMI tuning note: ring collectives on the 8-GPU full mesh are bound by a
# single 153 GB/s link, so larger buckets (fewer, bigger collectives) win.
# 5e8 elements (the reference default, = 1 GB bf16) is kept; it already
# amortizes well on xGMI and keeps peak memory bounded.
DEFAULT_BUCKET_SIZE = 500_000_000


class ConfigModel(BaseModel):
    model_config = {"extra": "forbid", "validate_assignment": True}


class CurriculumLearningConfig(ConfigModel):
    enabled: bool = False
    curriculum_type: str = "fixed_linear"
    min_difficulty: int = 1
    max_difficulty: int = 10
    schedule_config: dict = {}
    # difficulty source: "seqlen" derives per-sample difficulty from the
    # sample's length; "index" loads the DataAnalyzer's metric index file
    difficulty_metric: str = "seqlen"
    metric_path: Optional[str] = None


class RandomLTDConfig(ConfigModel):
    """Random layer-token-drop (reference data_pipeline/data_routing):
    middle layers run on a sampled token subset that grows on a schedule."""
    enabled: bool = False
    layers_attr: str = "model.layers"   # path to the decoder ModuleList
    skip_first: int = 1
    skip_last: int = 1
    min_value: int = 128                # starting kept-token count
    max_value: int = 4096               # full sequence by schedule end
    seq_per_step: int = 16
    total_ltd_steps: int = 1000


class EigenvalueConfig(ConfigModel):
    """Block Hessian eigenvalue estimation for MoQ quantization
    scheduling (reference runtime/eigenvalue.py + config key eigenvalue)."""
    enabled: bool = False
    verbose: bool = False
    max_iter: int = 100
    tol: float = 1e-2
    stability: float = 1e-6
    gas_boundary_resolution: int = 1
    layer_name: str = "layers"
    layer_num: int = 0


class ProgressiveLayerDropConfig(ConfigModel):
    """theta(t) keep-probability schedule (reference runtime/
    progressive_layer_drop.py; config key progressive_layer_drop)."""
    enabled: bool = False
    theta: float = 0.5
    gamma: float = 0.001


class DataEfficiencyConfig(ConfigModel):
    enabled: bool = False
    seed: int = 1234
    curriculum_learning: CurriculumLearningConfig = CurriculumLearningConfig()
    random_ltd: RandomLTDConfig = RandomLTDConfig()


class FP16Config(ConfigModel):
    enabled: bool = False
    loss_scale: float = 0.0  # 0 => dynamic
    initial_scale_power: int = 16
    loss_scale_window: int = 1000
    hysteresis: int = 2
    min_loss_scale: float = 1.0
    auto_cast: bool = False


class BF16Config(ConfigModel):
    enabled: bool = False


class OffloadOptimizerConfig(ConfigModel):
    device: str = "none"  # none | cpu | nvme
    pin_memory: bool = True
    nvme_path: str = "/tmp/zero_offload"
    buffer_count: int = 4
    fast_init: bool = False
    ratio: float = 1.0


class OffloadParamConfig(ConfigModel):
    device: str = "none"  # none | cpu | nvme
    pin_memory: bool = True
    nvme_path: str = "/tmp/zero_offload"
    buffer_count: int = 5
    buffer_size: int = 100_000_000
    max_in_cpu: int = 1_000_000_000


class ZeroConfig(ConfigModel):
    stage: int = 0
    reduce_bucket_size: int = DEFAULT_BUCKET_SIZE
    allgather_bucket_size: int = DEFAULT_BUCKET_SIZE
    overlap_comm: bool = True
    contiguous_gradients: bool = True
    reduce_scatter: bool = True
    offload_optimizer: OffloadOptimizerConfig = Field(default_factory=OffloadOptimizerConfig)
    offload_param: OffloadParamConfig = Field(default_factory=OffloadParamConfig)
    # stage-3 knobs
    stage3_prefetch_bucket_size: int = 50_000_000
    stage3_param_persistence_threshold: int = 100_000
    stage3_max_live_parameters: int = 1_000_000_000
    stage3_max_reuse_distance: int = 1_000_000_000
    stage3_gather_16bit_weights_on_model_save: bool = False
    sub_group_size: int = 1_000_000_000_000
    zero_hpz_partition_size: int = 1
    round_robin_gradients: bool = False
    # MiCS: shard params over sub-groups of this size, replicate across
    # sub-groups with a hierarchical gradient all-reduce (0 = off)
    mics_shard_size: int = 0
    # ZeRO++-style quantized weight all-gather (qwZ): ship int8 + group
    # scales over xGMI instead of bf16 (opt-in; changes forward numerics)
    zero_quantized_weights: bool = False
    zero_quantization_group_size: int = 2048
    # qgZ: quantized-gradient all-to-all reduce (two-level when
    # zero_hpz_partition_size groups are set — see runtime/zero/qgz.py)
    zero_quantized_gradients: bool = False
    # fp32 grad accumulation buffer (stage 1/2)
    fp32_grad_accum: bool = False


class OptimizerConfig(ConfigModel):
    type: str = "AdamW"
    params: Dict[str, Any] = Field(default_factory=dict)


class SchedulerConfig(ConfigModel):
    type: str = "WarmupLR"
    params: Dict[str, Any] = Field(default_factory=dict)


class ActivationCheckpointingConfig(ConfigModel):
    partition_activations: bool = False
    cpu_checkpointing: bool = False
    contiguous_memory_optimization: bool = False
    number_checkpoints: Optional[int] = None
    synchronize_checkpoint_boundary: bool = False
    profile: bool = False


class CommsLoggerConfig(ConfigModel):
    enabled: bool = False
    verbose: bool = False
    prof_all: bool = True
    debug: bool = False


class FlopsProfilerConfig(ConfigModel):
    enabled: bool = False
    profile_step: int = 1
    module_depth: int = -1
    top_modules: int = 1
    detailed: bool = True
    output_file: Optional[str] = None


class MonitorConfig(ConfigModel):
    enabled: bool = False
    tensorboard: Dict[str, Any] = Field(default_factory=dict)
    csv_monitor: Dict[str, Any] = Field(default_factory=dict)
    wandb: Dict[str, Any] = Field(default_factory=dict)
    comet: Dict[str, Any] = Field(default_factory=dict)


class MoEConfig(ConfigModel):
    enabled: bool = False
    ep_size: int = 1
    moe_param_group: bool = False


class PipelineConfig(ConfigModel):
    stages: int = 1
    partition_method: str = "parameters"
    activation_checkpoint_interval: int = 0


class TensorParallelConfig(ConfigModel):
    enabled: bool = False
    tp_size: int = 1


class AIOConfig(ConfigModel):
    block_size: int = 1048576
    queue_depth: int = 8
    thread_count: int = 1
    single_submit: bool = False
    overlap_events: bool = True


class Config:
    """Parsed + reconciled engine configuration.

    Accepts a dict or a path to a JSON file, mirroring
    ``DeepSpeedConfig(config, mpu)`` in the reference.
    """

    def __init__(self, config: Union[str, dict, None], world_size: int = 1):
        if config is None:
            config = {}
        if isinstance(config, str):
            with open(config) as f:
                config = json.load(f)
        self.raw: Dict[str, Any] = dict(config)
        g = self.raw.get

        self.train_batch_size = g("train_batch_size", None)
        self.train_micro_batch_size_per_gpu = g("train_micro_batch_size_per_gpu", None)
        self.gradient_accumulation_steps = g("gradient_accumulation_steps", None)
        self._reconcile_batch_sizes(world_size)

        self.steps_per_print = g("steps_per_print", 10)
        self.wall_clock_breakdown = g("wall_clock_breakdown", False)
        self.memory_breakdown = g("memory_breakdown", False)
        self.gradient_clipping = float(g("gradient_clipping", 0.0))
        self.prescale_gradients = g("prescale_gradients", False)
        self.gradient_predivide_factor = float(g("gradient_predivide_factor", 1.0))
        self.dump_state = g("dump_state", False)
        self.communication_data_type = g("communication_data_type", None)
        self.seq_parallel_communication_data_type = g(
            "seq_parallel_communication_data_type", None)
        self.sparse_gradients_enabled = g("sparse_gradients", False)
        # reference top-level "sparse_attention" block ({"mode": "fixed",
        # "block": 16, ...}); consumed via build_sparse_attention()
        self.sparse_attention = g("sparse_attention", None)
        self.zero_allow_untested_optimizer = g("zero_allow_untested_optimizer", True)

        self.fp16 = FP16Config(**g("fp16", {}))
        self.bf16 = BF16Config(**g("bf16", {}))
        self.zero = ZeroConfig(**g("zero_optimization", {}))
        self.optimizer: Optional[OptimizerConfig] = (
            OptimizerConfig(**g("optimizer")) if g("optimizer") else None)
        self.scheduler: Optional[SchedulerConfig] = (
            SchedulerConfig(**g("scheduler")) if g("scheduler") else None)
        self.activation_checkpointing = ActivationCheckpointingConfig(
            **g("activation_checkpointing", {}))
        self.comms_logger = CommsLoggerConfig(**g("comms_logger", {}))
        self.flops_profiler = FlopsProfilerConfig(**g("flops_profiler", {}))
        # reference accepts the writer blocks at the top level of ds_config
        # (tensorboard/wandb/comet/csv_monitor) as well as grouped
        mon = dict(g("monitor_config", g("monitor", {})))
        for k in ("tensorboard", "wandb", "comet", "csv_monitor"):
            blk = g(k, None)
            if blk is not None and k not in mon:
                mon[k] = blk
        # any enabled writer block activates the monitor (reference
        # monitor_config semantics: the writers carry their own enables)
        if not mon.get("enabled") and any(
                isinstance(mon.get(k), dict) and mon[k].get("enabled")
                for k in ("tensorboard", "wandb", "comet", "csv_monitor")):
            mon["enabled"] = True
        self.monitor = MonitorConfig(**mon)
        self.moe = MoEConfig(**g("moe", {}))
        self.pipeline = PipelineConfig(**g("pipeline", {}))
        self.tensor_parallel = TensorParallelConfig(**g("tensor_parallel", {}))
        self.aio = AIOConfig(**g("aio", {}))
        self.data_efficiency = DataEfficiencyConfig(
            **g("data_efficiency", {}))
        self.progressive_layer_drop = ProgressiveLayerDropConfig(
            **g("progressive_layer_drop", {}))
        self.eigenvalue = EigenvalueConfig(**g("eigenvalue", {}))

        self.data_types_grad_accum_dtype = (
            g("data_types", {}).get("grad_accum_dtype", None))

        if self.fp16.enabled and self.bf16.enabled:
            raise ValueError("fp16 and bf16 cannot both be enabled")

    # -- batch reconciliation (reference runtime/config.py train-batch logic) --
    def _reconcile_batch_sizes(self, world_size: int) -> None:
        tb, mb, gas = (self.train_batch_size, self.train_micro_batch_size_per_gpu,
                       self.gradient_accumulation_steps)
        ws = max(world_size, 1)
        if tb is not None and mb is not None and gas is not None:
            if tb != mb * gas * ws:
                raise ValueError(
                    f"train_batch_size {tb} != micro_batch {mb} * gas {gas} * world {ws}")
        elif tb is not None and mb is not None:
            gas = tb // (mb * ws)
            if gas * mb * ws != tb:
                raise ValueError("train_batch_size not divisible by micro_batch*world")
        elif tb is not None and gas is not None:
            mb = tb // (gas * ws)
            if mb * gas * ws != tb:
                raise ValueError("train_batch_size not divisible by gas*world")
        elif mb is not None and gas is not None:
            tb = mb * gas * ws
        elif tb is not None:
            mb = max(tb // ws, 1)
            gas = tb // (mb * ws)
            if mb * gas * ws != tb:
                raise ValueError(f"cannot infer micro batch from train_batch_size {tb}")
        elif mb is not None:
            gas = 1
            tb = mb * ws
        else:
            mb, gas = 1, 1
            tb = ws
        self.train_batch_size, self.train_micro_batch_size_per_gpu = tb, mb
        self.gradient_accumulation_steps = gas

    @property
    def dtype(self):
        import torch
        if self.bf16.enabled:
            return torch.bfloat16
        if self.fp16.enabled:
            return torch.float16
        return torch.float32

    def print_config(self):
        logger.info(json.dumps(self.raw, indent=2, sort_keys=True, default=str))
