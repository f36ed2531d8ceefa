"""Core training engine.

Capability parity with the reference's ``deepspeed/runtime/engine.py``
(DeepSpeedEngine :184 — forward :1926 / backward :2085 / step :2282 /
save_checkpoint :3218 / load_checkpoint :2872 / allreduce fallback :2611),
re-designed for one-process-per-GPU over RCCL/xGMI on MI355X.

Precision policy (simpler than the reference's three optimizer stacks):
* fp32 + ZeRO stage 0 -> plain DDP-style bucketed all-reduce + user optimizer
* bf16/fp16 at stage 0/1 -> ZeRO-1 partitioned fp32-master optimizer
  (this is what the reference's BF16_Optimizer does)
* stage 2 -> ZeRO-2 (gradient partitioning via reduce-scatter)
* stage 3 -> ZeRO-3 (parameter partitioning; see zero/stage3.py)
"""

import os
from typing import Optional

import torch

from .. import accel
from .. import comm as dist
from ..config import Config
from ..parallel import groups
from ..utils.logging import log_dist, logger
from ..utils.timer import (SynchronizedWallClockTimer, ThroughputTimer,
                           FORWARD_GLOBAL_TIMER, BACKWARD_GLOBAL_TIMER,
                           STEP_GLOBAL_TIMER)
from . import lr_schedules
from .fp16.loss_scaler import create_loss_scaler, LossScaler
from .utils import (DummyOptim, get_global_norm_of_tensors,
                    clip_tensors_by_global_norm, see_memory_usage)
from .zero.stage12 import ZeroStage12Optimizer
from ..utils.nvtx import instrument_w_nvtx

MEMORY_OPT_ALLREDUCE_SIZE = 500_000_000


class Engine(torch.nn.Module):
    """Training wrapper returned by :func:`deepspeed_amd.initialize`."""

    def __init__(self,
                 model: torch.nn.Module,
                 optimizer: Optional[torch.optim.Optimizer] = None,
                 model_parameters=None,
                 lr_scheduler=None,
                 config: Optional[Config] = None,
                 mpu=None,
                 dont_change_device: bool = False):
        super().__init__()
        self.module = model
        self.client_optimizer = optimizer
        self.client_lr_scheduler = lr_scheduler
        self.config = config if config is not None else Config(None)
        self.mpu = mpu
        self.global_steps = 0
        self.global_samples = 0
        self.micro_steps = 0
        self.skipped_steps = 0
        self.gradient_accumulation_steps = self.config.gradient_accumulation_steps
        self.train_micro_batch_size_per_gpu = self.config.train_micro_batch_size_per_gpu
        self.train_batch_size = self.config.train_batch_size
        self._is_gradient_accumulation_boundary = True
        self.inside_no_sync = False

        if not dist.is_initialized():
            dist.init_distributed()
        if mpu is not None:
            groups.set_mpu(mpu)
        self.dp_group = groups.get_data_parallel_group()
        self.dp_world_size = dist.get_world_size(self.dp_group)
        self.seq_parallel_world_size = groups.get_sequence_parallel_world_size()

        self.device = accel.current_device() if dont_change_device else \
            accel.device_for(accel.local_rank_from_env())
        self.local_rank = accel.local_rank_from_env()
        self.global_rank = dist.get_rank()
        self.world_size = dist.get_world_size()

        self.timers = SynchronizedWallClockTimer()
        self.tput_timer = ThroughputTimer(
            batch_size=self.train_batch_size,
            steps_per_output=self.config.steps_per_print)
        self.wall_clock_breakdown = self.config.wall_clock_breakdown

        if self.config.comms_logger.enabled:
            dist.configure_comms_logger(
                enabled=True,
                verbose=self.config.comms_logger.verbose,
                prof_all=self.config.comms_logger.prof_all,
                debug=self.config.comms_logger.debug)

        self.monitor = None
        if self.config.monitor.enabled:
            from ..monitor.monitor import MonitorMaster
            self.monitor = MonitorMaster(self.config.monitor)

        from .activation_checkpointing import configure as _ac_configure
        _ac_configure(self.config.activation_checkpointing, mpu=self.mpu)

        self.flops_profiler = None
        if self.config.flops_profiler.enabled:
            from ..profiling.flops_profiler import FlopsProfiler
            self.flops_profiler = FlopsProfiler(self.module)

        self.zero_stage = self.config.zero.stage
        self.dtype = self.config.dtype

        # MoE: create EP / expert-DP groups before optimizer partitioning
        from ..moe.layer import MoE, has_moe_layers
        self.has_moe_layers = has_moe_layers(self.module)
        if self.has_moe_layers:
            for m in self.module.modules():
                if isinstance(m, MoE):
                    m.set_deepspeed_parallelism()
            if self.zero_stage == 3:
                raise ValueError("MoE expert parallelism composes with ZeRO "
                                 "stages 0-2 (expert params are partitioned "
                                 "over the expert-DP group, not ZeRO-3)")

        from .checkpoint_engine import create_checkpoint_engine
        self.checkpoint_engine = create_checkpoint_engine(
            self.config.raw.get("checkpoint", {}).get("engine", "torch"))

        # random-LTD (reference data_routing/basic_layer.py): wrap middle
        # decoder layers BEFORE the optimizer partitions params (wrapping
        # preserves param identity, but ZeRO-3 records module units)
        self.random_ltd_scheduler = None
        ltd = self.config.data_efficiency.random_ltd
        if self.config.data_efficiency.enabled and ltd.enabled:
            from .data_pipeline.random_ltd import (RandomLTDScheduler,
                                                   convert_to_random_ltd)
            obj = self.module
            for part in ltd.layers_attr.split("."):
                obj = getattr(obj, part)
            self.random_ltd_scheduler = RandomLTDScheduler(
                total_layers=len(obj),
                random_ltd_layer_num=len(obj) - ltd.skip_first -
                ltd.skip_last,
                start_seq=ltd.min_value, max_seq=ltd.max_value,
                step_size=ltd.seq_per_step,
                schedule_steps=ltd.total_ltd_steps)
            n = convert_to_random_ltd(self.module, ltd.layers_attr,
                                      self.random_ltd_scheduler,
                                      skip_first=ltd.skip_first,
                                      skip_last=ltd.skip_last)
            log_dist(f"random-LTD: wrapped {n} layers, kept tokens "
                     f"{ltd.min_value}->{ltd.max_value}")

        self.eigenvalue = None
        ev = self.config.eigenvalue
        if ev.enabled:
            from .eigenvalue import Eigenvalue
            self.eigenvalue = Eigenvalue(
                verbose=ev.verbose, max_iter=ev.max_iter, tol=ev.tol,
                stability=ev.stability,
                gas_boundary_resolution=ev.gas_boundary_resolution,
                layer_name=ev.layer_name, layer_num=ev.layer_num)
            # tag the blocks whose sensitivity MoQ schedules by: children
            # of the ModuleList/attr named layer_name
            for name, mod in self.module.named_modules():
                parts = name.split(".")
                if len(parts) >= 2 and parts[-2] == ev.layer_name and \
                        parts[-1].isdigit():
                    mod._deepspeed_eigenvalue_block = True

        self.progressive_layer_drop = None
        if self.config.progressive_layer_drop.enabled:
            from .progressive_layer_drop import ProgressiveLayerDrop
            self.progressive_layer_drop = ProgressiveLayerDrop(
                theta=self.config.progressive_layer_drop.theta,
                gamma=self.config.progressive_layer_drop.gamma)

        self._configure_distributed_model(dont_change_device)
        self._configure_optimizer(model_parameters)
        self._configure_lr_scheduler()

        see_memory_usage("engine initialized", force=self.config.memory_breakdown)

    # ------------------------------------------------------------------ setup

    def _configure_distributed_model(self, dont_change_device):
        if self.zero_stage < 3:
            if self.dtype != torch.float32:
                self.module.to(self.dtype)
            if not dont_change_device:
                self.module.to(self.device)
            self._broadcast_model()
        else:
            # ZeRO-3: params may already be partitioned (zero.Init); the
            # stage-3 optimizer handles placement and bcast. Meta-init'd
            # modules (zero.Init(remote_device="meta")) must not be moved —
            # stage 3 materializes them unit by unit at partition time.
            if not dont_change_device and \
                    not any(p.is_meta for p in self.module.parameters()):
                self.module.to(self.device)

    def _broadcast_model(self):
        """Replicate rank-0 weights across the DP group; expert params are
        broadcast over their expert-DP group instead, since each EP rank owns
        different experts (reference :1183)."""
        if self.dp_world_size == 1:
            return
        from ..parallel import groups as pgroups
        # src must be the group's OWN first rank: with TP/pipe grids the DP
        # group need not contain global rank 0
        dp_src = torch.distributed.get_global_rank(self.dp_group, 0) \
            if self.dp_group is not None and \
            self.dp_group is not torch.distributed.group.WORLD else 0
        for p in self.module.parameters():
            if not torch.is_tensor(p):
                continue
            if getattr(p, "allreduce", True) is False:
                g = pgroups.get_expert_data_parallel_group(p.group_name)
                if dist.get_world_size(g) > 1:
                    src = torch.distributed.get_global_rank(g, 0)
                    dist.broadcast(p.data, src=src, group=g)
            else:
                dist.broadcast(p.data, src=dp_src, group=self.dp_group)
        for b in self.module.buffers():
            if torch.is_tensor(b) and b.numel() > 0 and b.dtype.is_floating_point:
                dist.broadcast(b.data, src=dp_src, group=self.dp_group)

    def _configure_basic_optimizer(self, model_parameters):
        cfg = self.config.optimizer
        if cfg is None:
            return None
        params = dict(cfg.params)
        name = cfg.type.lower()
        if model_parameters is None:
            model_parameters = [p for p in self.module.parameters() if p.requires_grad]
        if isinstance(model_parameters, list) and not model_parameters:
            # e.g. a parameterless pipeline stage (activation-only layers)
            log_dist("no trainable parameters on this rank - DummyOptim")
            return DummyOptim([])
        if self.has_moe_layers:
            from ..moe.layer import \
                split_params_into_different_moe_groups_for_optimizer
            if isinstance(model_parameters, list) and model_parameters and \
                    not isinstance(model_parameters[0], dict):
                model_parameters = [{"params": model_parameters}]
            model_parameters = \
                split_params_into_different_moe_groups_for_optimizer(model_parameters)
        if name in ("adam", "adamw", "fusedadam"):
            adam_w = params.pop("adam_w_mode", name != "adam")
            if name == "adamw":
                adam_w = True
            from ..ops.adam import FusedAdam
            return FusedAdam(model_parameters, adam_w_mode=adam_w, **params)
        if name in ("deepspeedcpuadam", "cpuadam", "cpu_adam"):
            from ..ops.adam import DeepSpeedCPUAdam
            adam_w = params.pop("adam_w_mode", True)
            return DeepSpeedCPUAdam(model_parameters, adam_w_mode=adam_w,
                                    **params)
        if name == "sgd":
            return torch.optim.SGD(model_parameters, **params)
        if name == "lion":
            from ..ops.lion import Lion
            return Lion(model_parameters, **params)
        if name in ("lamb", "fusedlamb"):
            from ..ops.lamb import FusedLamb
            return FusedLamb(model_parameters, **params)
        if name == "adagrad":
            return torch.optim.Adagrad(model_parameters, **params)
        if name in ("onebitadam", "onebit_adam"):
            from .fp16.onebit import OnebitAdam
            return OnebitAdam(model_parameters, **params)
        if name in ("onebitlamb", "onebit_lamb"):
            from .fp16.onebit import OnebitLamb
            return OnebitLamb(model_parameters, **params)
        if name in ("zerooneadam", "zero_one_adam"):
            from .fp16.onebit import ZeroOneAdam
            return ZeroOneAdam(model_parameters, **params)
        raise ValueError(f"unsupported optimizer type {cfg.type}")

    def _configure_optimizer(self, model_parameters):
        basic = self.client_optimizer or self._configure_basic_optimizer(model_parameters)
        self.basic_optimizer = basic
        if basic is None:
            self.optimizer = None
            return

        if isinstance(basic, DummyOptim):
            self.optimizer = basic
            return

        stage = self.zero_stage
        mixed = self.dtype in (torch.float16, torch.bfloat16)
        if stage == 3:
            from .zero.stage3 import ZeroStage3Optimizer
            self.optimizer = ZeroStage3Optimizer(
                module=self.module,
                init_optimizer=basic,
                dp_group=self.dp_group,
                config=self.config,
                loss_scaler=self._make_loss_scaler(),
            )
        elif stage in (1, 2) or mixed:
            eff_stage = stage if stage in (1, 2) else 1
            offload = self.config.zero.offload_optimizer.device == "cpu"
            self.optimizer = ZeroStage12Optimizer(
                init_optimizer=basic,
                stage=eff_stage,
                dp_group=self.dp_group,
                reduce_bucket_size=self.config.zero.reduce_bucket_size,
                allgather_bucket_size=self.config.zero.allgather_bucket_size,
                overlap_comm=self.config.zero.overlap_comm,
                clip_grad=self.config.gradient_clipping,
                loss_scaler=self._make_loss_scaler(),
                communication_dtype=self._comm_dtype(),
                gradient_predivide_factor=self.config.gradient_predivide_factor,
                cpu_offload=offload,
                offload_pin_memory=self.config.zero.offload_optimizer.pin_memory,
                grad_accum_dtype=(torch.float32
                                  if self.config.zero.fp32_grad_accum
                                  else None),
                mpu=self.mpu,
            )
        else:
            self.optimizer = basic  # fp32 stage 0: engine handles allreduce
        if hasattr(self.optimizer, "annotate_param_names"):
            # names feed the universal-checkpoint layout manifest
            self.optimizer.annotate_param_names(self.module)

    def _make_loss_scaler(self):
        if self.dtype == torch.float16:
            return create_loss_scaler(self.config.fp16)
        return LossScaler(1.0)

    def _comm_dtype(self):
        name = self.config.communication_data_type
        if name is None:
            return None
        return {"fp32": torch.float32, "fp16": torch.float16,
                "bf16": torch.bfloat16}[name]

    def _configure_lr_scheduler(self):
        if self.client_lr_scheduler is not None:
            self.lr_scheduler = self.client_lr_scheduler
            return
        cfg = self.config.scheduler
        if cfg is None or self.optimizer is None:
            self.lr_scheduler = None
            return
        self.lr_scheduler = lr_schedules.get_scheduler(cfg.type, self.optimizer,
                                                       cfg.params)

    # -------------------------------------------------------------- accessors

    def get_lr(self):
        if self.optimizer is None:
            return []
        return [g["lr"] for g in self.optimizer.param_groups]

    def get_global_grad_norm(self):
        if hasattr(self.optimizer, "get_global_grad_norm"):
            return self.optimizer.get_global_grad_norm()
        return getattr(self, "_fallback_grad_norm", 0.0)

    # ---- reference-parity config accessors (engine.py:588-1010). The
    # reference exposes these as METHODS; frameworks built on DeepSpeed
    # (Megatron-DeepSpeed, trainer integrations) call them, so they are
    # methods here too. train_batch_size / gradient_accumulation_steps /
    # train_micro_batch_size_per_gpu remain plain attributes (documented
    # delta in docs/migrating_from_deepspeed.md).

    def fp16_enabled(self):
        return self.config.fp16.enabled

    def bfloat16_enabled(self):
        return self.config.bf16.enabled

    def fp16_auto_cast(self):
        return self.config.fp16.auto_cast

    def amp_enabled(self):
        return False  # torch.amp path not used; bf16/fp16 policies instead

    def dynamic_loss_scale(self):
        return self.config.fp16.loss_scale == 0

    def initial_dynamic_scale(self):
        return 2.0 ** self.config.fp16.initial_scale_power

    def zero_optimization_stage(self):
        return self.zero_stage

    def zero_optimization_partition_gradients(self):
        return self.zero_stage >= 2

    def zero_optimization_partition_weights(self):
        return self.zero_stage >= 3

    def zero_overlap_comm(self):
        return self.config.zero.overlap_comm

    def zero_reduce_bucket_size(self):
        return self.config.zero.reduce_bucket_size

    def zero_allgather_bucket_size(self):
        return self.config.zero.allgather_bucket_size

    def zero_offload_optimizer(self):
        return self.config.zero.offload_optimizer

    def zero_offload_param(self):
        return self.config.zero.offload_param

    def zero_cpu_offload(self):
        return self.config.zero.offload_optimizer.device in ("cpu", "nvme")

    def zero_prefetch_bucket_size(self):
        return self.config.zero.stage3_prefetch_bucket_size

    def zero_param_persistence_threshold(self):
        return self.config.zero.stage3_param_persistence_threshold

    def zero_max_live_parameters(self):
        return self.config.zero.stage3_max_live_parameters

    def zero_max_reuse_distance(self):
        return self.config.zero.stage3_max_reuse_distance

    def zero_gather_16bit_weights_on_model_save(self):
        return self.config.zero.stage3_gather_16bit_weights_on_model_save

    def zero_hpz_partition_size(self):
        return self.config.zero.zero_hpz_partition_size

    def zero_quantized_weights(self):
        return self.config.zero.zero_quantized_weights

    def zero_quantized_gradients(self):
        return self.config.zero.zero_quantized_gradients

    def mics_shard_size(self):
        return self.config.zero.mics_shard_size

    def gradient_clipping(self):
        return self.config.gradient_clipping

    def gradient_predivide_factor(self):
        return self.config.gradient_predivide_factor

    def steps_per_print(self):
        return self.config.steps_per_print

    def memory_breakdown(self):
        return self.config.memory_breakdown

    def sparse_gradients_enabled(self):
        return self.config.sparse_gradients_enabled

    def optimizer_name(self):
        if self.client_optimizer is not None:
            return type(self.client_optimizer).__name__
        return self.config.optimizer.type

    def optimizer_params(self):
        return self.config.optimizer.params

    def scheduler_name(self):
        return self.config.scheduler.type if self.config.scheduler else None

    def scheduler_params(self):
        return self.config.scheduler.params if self.config.scheduler else None

    def flops_profiler_enabled(self):
        return self.config.flops_profiler.enabled

    def curriculum_learning_enabled(self):
        return self.config.data_efficiency.curriculum_learning.enabled

    def data_efficiency_enabled(self):
        return self.config.data_efficiency.enabled

    def random_ltd_enabled(self):
        return self.random_ltd_scheduler is not None

    def pld_enabled(self):
        return self.progressive_layer_drop is not None

    def pld_theta(self):
        return self.get_pld_theta()

    def aio_config(self):
        return self.config.aio

    def communication_data_type(self):
        return self._comm_dtype() or self.dtype

    def get_batch_info(self):
        return (self.train_batch_size, self.train_micro_batch_size_per_gpu,
                self.gradient_accumulation_steps)

    def get_data_types(self):
        return (self.dtype, torch.float32)

    def was_step_applied(self) -> bool:
        """False when the last step was skipped (fp16 overflow)."""
        return not bool(getattr(self.optimizer, "overflow", False))

    def zero_grad(self):
        with torch.no_grad():
            # grads are VIEWS into the ZeRO flat buffers; zero in place
            # (detach_/None would break the view scheme)
            for p in self.module.parameters():
                if p.grad is not None:
                    p.grad.zero_()

    def empty_partition_cache(self):
        """Release every gathered ZeRO-3 full buffer (reference
        engine.empty_partition_cache) and return cached HBM to the pool."""
        opt = self.optimizer
        if hasattr(opt, "units") and hasattr(opt, "_release"):
            for u in opt.units:
                if not u.persist:
                    opt._release(u)
        if accel.available():
            torch.cuda.empty_cache()

    def save_fp16_model(self, save_dir, save_filename="pytorch_model.bin"):
        """Reference-compat alias of save_16bit_model."""
        return self.save_16bit_model(save_dir, save_filename)

    def load_module_state_dict(self, state_dict, strict=True):
        self.module.load_state_dict(state_dict, strict=strict)

    def destroy(self):
        """Detach grad hooks and break engine<->optimizer cycles so the
        model can be reused outside the engine."""
        opt = self.optimizer
        for attr in ("_grad_hooks", "_hooks"):
            for h in getattr(opt, attr, []) or []:
                try:
                    h.remove()
                except Exception:
                    pass
        self.optimizer = None

    def set_custom_curriculum_learning_schedule(self, fn):
        """Replace the pacing function of the curriculum scheduler
        (curriculum_type "custom"): fn(global_steps) -> difficulty
        (reference engine.set_custom_curriculum_learning_schedule)."""
        sched = getattr(self, "curriculum_scheduler", None)
        assert sched is not None, \
            "build the dataloader via engine.deepspeed_io with " \
            "curriculum_learning enabled first"
        sched.set_custom_get_difficulty(fn)

    def set_data_post_process_func(self, fn):
        """Post-process hook fn(batch, difficulty) applied to every batch
        the curriculum dataloader yields (reference
        engine.set_data_post_process_func — e.g. truncate samples to the
        current difficulty). Call BEFORE engine.deepspeed_io."""
        self._data_post_process_func = fn

    def get_mom(self):
        """Momentum / betas of the first param group (reference
        engine.get_mom:2474)."""
        if self.optimizer is None:
            return []
        key = "momentum" if "momentum" in self.optimizer.param_groups[0] \
            else "betas"
        return [g.get(key) for g in self.optimizer.param_groups]

    def get_pld_theta(self):
        return (self.progressive_layer_drop.get_theta()
                if self.progressive_layer_drop is not None else None)

    def set_train_batch_size(self, train_batch_size: int):
        """Change the global batch size by adjusting gradient accumulation
        (micro-batch size and DP world are fixed; reference
        engine.set_train_batch_size:524). Used by the autotuner and
        curriculum schedules."""
        denom = self.train_micro_batch_size_per_gpu * self.dp_world_size
        if train_batch_size % denom != 0:
            raise ValueError(
                f"train_batch_size {train_batch_size} not divisible by "
                f"micro_batch x dp = {denom}")
        self.gradient_accumulation_steps = train_batch_size // denom
        self.train_batch_size = train_batch_size
        self.config.gradient_accumulation_steps = \
            self.gradient_accumulation_steps
        self.config.train_batch_size = train_batch_size
        if self.tput_timer:
            self.tput_timer.batch_size = train_batch_size

    def set_train_micro_batch_size(self, micro_batch_size: int):
        """Change the per-GPU micro batch; gradient accumulation steps stay
        fixed, so the global batch scales with it."""
        self.train_micro_batch_size_per_gpu = micro_batch_size
        self.train_batch_size = micro_batch_size * \
            self.gradient_accumulation_steps * self.dp_world_size
        self.config.train_micro_batch_size_per_gpu = micro_batch_size
        self.config.train_batch_size = self.train_batch_size
        if self.tput_timer:
            self.tput_timer.batch_size = self.train_batch_size

    @property
    def loss_scale(self):
        if hasattr(self.optimizer, "loss_scale"):
            return self.optimizer.loss_scale
        return 1.0

    def zero_optimization(self):
        return self.zero_stage > 0

    def is_gradient_accumulation_boundary(self) -> bool:
        return self._is_gradient_accumulation_boundary

    def set_gradient_accumulation_boundary(self, is_boundary: bool):
        self._is_gradient_accumulation_boundary = is_boundary
        if hasattr(self.optimizer, "is_gradient_accumulation_boundary"):
            self.optimizer.is_gradient_accumulation_boundary = is_boundary

    def train(self, mode=True):
        self.module.train(mode)
        return self

    def eval(self):
        self.module.eval()
        return self

    # ------------------------------------------------------------ hot path

    @instrument_w_nvtx
    def forward(self, *inputs, **kwargs):
        if self.config.fp16.enabled and self.config.fp16.auto_cast:
            # reference fp16 auto_cast: float inputs arrive fp32, cast once
            # at the engine boundary instead of inside every module
            inputs = tuple(t.to(self.dtype) if torch.is_tensor(t)
                           and t.is_floating_point() else t for t in inputs)
            kwargs = {k: (v.to(self.dtype) if torch.is_tensor(v)
                          and v.is_floating_point() else v)
                      for k, v in kwargs.items()}
        if self.wall_clock_breakdown:
            self.timers(FORWARD_GLOBAL_TIMER).start()
        if self.flops_profiler is not None and \
                self.global_steps == self.config.flops_profiler.profile_step and \
                self.micro_steps % self.gradient_accumulation_steps == 0:
            self.flops_profiler.start_profile(ignore_list=None)
        loss = self.module(*inputs, **kwargs)
        if self.flops_profiler is not None and \
                self.global_steps == self.config.flops_profiler.profile_step and \
                self.micro_steps % self.gradient_accumulation_steps == 0:
            self.flops_profiler.stop_profile()
            if self.global_rank == 0:
                self.flops_profiler.print_model_profile(
                    profile_step=self.global_steps,
                    module_depth=self.config.flops_profiler.module_depth,
                    top_modules=self.config.flops_profiler.top_modules,
                    detailed=self.config.flops_profiler.detailed,
                    output_file=self.config.flops_profiler.output_file)
            self.flops_profiler.end_profile()
        if self.wall_clock_breakdown:
            self.timers(FORWARD_GLOBAL_TIMER).stop()
        return loss

    @instrument_w_nvtx
    def backward(self, loss, retain_graph=False, scale_wrt_gas=True):
        boundary = (self.micro_steps + 1) % self.gradient_accumulation_steps == 0 \
            and not self.inside_no_sync
        self.set_gradient_accumulation_boundary(boundary)

        if self.wall_clock_breakdown:
            self.timers(BACKWARD_GLOBAL_TIMER).start()
        if self.tput_timer:
            self.tput_timer.start()

        if self.gradient_accumulation_steps > 1 and scale_wrt_gas:
            loss = loss / self.gradient_accumulation_steps
        # Ulysses note: NO extra loss scaling for SP — ZeRO averages grads
        # over the full DPxSP mesh (see parallel/groups.py), which already
        # yields the gradient of the global token-mean loss.

        # keep as a tensor: .item() here would force a device sync every
        # microstep and stall CPU run-ahead; materialized only when the
        # monitor actually writes
        self._last_loss = loss.detach()
        if hasattr(self.optimizer, "backward"):
            self.optimizer.backward(loss, retain_graph=retain_graph)
        else:
            loss.backward(retain_graph=retain_graph)

        if boundary:
            self.allreduce_gradients()

        if self.wall_clock_breakdown:
            self.timers(BACKWARD_GLOBAL_TIMER).stop()
        self.micro_steps += 1
        self.global_samples += self.train_micro_batch_size_per_gpu
        return loss

    def allreduce_gradients(self):
        """At the GAS boundary: flush the ZeRO reducer, or run the bucketed
        DDP-fallback all-reduce for stage-0 fp32 (reference :2611)."""
        if hasattr(self.optimizer, "reduce_gradients"):
            self.optimizer.reduce_gradients()
        elif self.dp_world_size > 1:
            self._buffered_allreduce_fallback()

    @torch.no_grad()
    def _sparse_allreduce(self, grad):
        """Average a sparse gradient (sparse embeddings) across DP by
        all-gathering indices+values (reference engine.py:2627)."""
        grad = grad.coalesce()
        parts = [None] * self.dp_world_size
        dist.all_gather_object(parts, (grad.indices().cpu(),
                                       grad.values().cpu()),
                               group=self.dp_group)
        idx = torch.cat([p[0] for p in parts], dim=1).to(grad.device)
        val = torch.cat([p[1] for p in parts], dim=0).to(grad.device)
        return torch.sparse_coo_tensor(idx, val / self.dp_world_size,
                                       grad.shape).coalesce()

    @torch.no_grad()
    def _buffered_allreduce_fallback(self, elements_per_buffer=MEMORY_OPT_ALLREDUCE_SIZE):
        for p in self.module.parameters():
            if p.grad is not None and p.grad.is_sparse:
                p.grad = self._sparse_allreduce(p.grad)
        grads = [p.grad for p in self.module.parameters()
                 if p.grad is not None and not p.grad.is_sparse]
        bucket, bucket_elems = [], 0
        from torch._utils import _flatten_dense_tensors, _unflatten_dense_tensors

        def flush():
            nonlocal bucket, bucket_elems
            if not bucket:
                return
            flat = _flatten_dense_tensors(bucket)
            flat.div_(self.dp_world_size)
            dist.all_reduce(flat, group=self.dp_group)
            for g, synced in zip(bucket, _unflatten_dense_tensors(flat, bucket)):
                g.copy_(synced)
            bucket, bucket_elems = [], 0

        for g in grads:
            bucket.append(g)
            bucket_elems += g.numel()
            if bucket_elems >= elements_per_buffer:
                flush()
        flush()

    @instrument_w_nvtx
    def step(self, lr_kwargs=None):
        if not self._is_gradient_accumulation_boundary:
            return
        if self.wall_clock_breakdown:
            self.timers(STEP_GLOBAL_TIMER).start()

        if self.optimizer is not None and not isinstance(self.optimizer, DummyOptim):
            if isinstance(self.optimizer, torch.optim.Optimizer) and \
                    not hasattr(self.optimizer, "reduce_gradients"):
                # plain fp32 stage-0 path: clip + step + zero
                if self.config.gradient_clipping > 0:
                    params = [p for p in self.module.parameters() if p.grad is not None]
                    norm = get_global_norm_of_tensors(
                        [p.grad for p in params], group=None)
                    self._fallback_grad_norm = float(norm)
                    clip_tensors_by_global_norm([p.grad for p in params],
                                                self.config.gradient_clipping, norm)
                self.optimizer.step()
                self.optimizer.zero_grad()
            else:
                self.optimizer.step()
                self.optimizer.zero_grad()

        overflow = bool(getattr(self.optimizer, "overflow", False))
        if overflow:
            self.skipped_steps += 1
        elif self.lr_scheduler is not None:
            try:
                self.lr_scheduler.step(**(lr_kwargs or {}))
            except TypeError:
                self.lr_scheduler.step()

        self.global_steps += 1
        if self.random_ltd_scheduler is not None:
            self.random_ltd_scheduler.update_seq(self.global_steps)
        if self.progressive_layer_drop is not None:
            self.progressive_layer_drop.update_state(self.global_steps)
        if self.tput_timer:
            self.tput_timer.stop(global_step=True)
        if self.monitor is not None and self.global_rank == 0:
            events = [("Train/lr", self.get_lr()[0] if self.get_lr() else 0.0,
                       self.global_steps),
                      ("Train/loss_scale", float(self.loss_scale),
                       self.global_steps)]
            if getattr(self, "_last_loss", None) is not None:
                events.append(("Train/loss", float(self._last_loss),
                               self.global_steps))
            if self.tput_timer and self.tput_timer.avg_samples_per_sec():
                events.append(("Train/samples_per_sec",
                               self.tput_timer.avg_samples_per_sec(),
                               self.global_steps))
            if getattr(self, "curriculum_scheduler", None) is not None:
                events.append(
                    ("Train/curriculum_difficulty",
                     self.curriculum_scheduler.get_current_difficulty(),
                     self.global_steps))
            if self.has_moe_layers and hasattr(self.module, "aux_loss"):
                try:
                    events.append(("Train/moe_aux_loss",
                                   float(self.module.aux_loss()),
                                   self.global_steps))
                except Exception:
                    pass
            self.monitor.write_events(events)
        if self.wall_clock_breakdown:
            self.timers(STEP_GLOBAL_TIMER).stop()
            if self.global_steps % self.config.steps_per_print == 0:
                self.timers.log([FORWARD_GLOBAL_TIMER, BACKWARD_GLOBAL_TIMER,
                                 STEP_GLOBAL_TIMER])
        self.set_gradient_accumulation_boundary(True)

    class _NoSync:
        def __init__(self, engine):
            self.engine = engine

        def __enter__(self):
            self.engine.inside_no_sync = True

        def __exit__(self, *a):
            self.engine.inside_no_sync = False
            return False

    def no_sync(self):
        """Context manager: disable gradient reduction inside (reference :2065)."""
        return Engine._NoSync(self)

    # ------------------------------------------------------ state offloading

    def _optimizer_state_tensors(self):
        """Yield (container, key, tensor) for every optimizer-state tensor
        (ZeRO flat masters/accumulators + inner Adam moments)."""
        opt = self.optimizer
        if opt is None:
            return
        for attr in ("group_masters", "group_owned_grads"):
            lst = getattr(opt, attr, None)
            if lst is not None:
                for i, t in enumerate(lst):
                    if torch.is_tensor(t) and t.numel():
                        yield lst, i, t
        inner = getattr(opt, "optimizer", opt)
        for state in getattr(inner, "state", {}).values():
            for k, v in state.items():
                if torch.is_tensor(v) and v.numel():
                    yield state, k, v

    @torch.no_grad()
    def offload_states(self, include=None, device="cpu", pin_memory=True,
                       non_blocking=False):
        """Move optimizer states to host DRAM to free HBM between phases
        (reference engine.py:3844) — e.g. before a long generate() in RLHF."""
        moved = 0
        for container, key, t in list(self._optimizer_state_tensors()):
            if not t.is_cuda:
                continue
            host = torch.empty(t.shape, dtype=t.dtype, device=device)
            if pin_memory and device == "cpu" and accel.available():
                host = host.pin_memory()
            host.copy_(t, non_blocking=non_blocking)
            if isinstance(t, torch.nn.Parameter):
                # inner-optimizer params are referenced from param_groups
                # and as state keys: swap storage, keep identity
                t.data = host
            else:
                container[key] = host
            moved += t.numel() * t.element_size()
        if accel.available():
            accel.synchronize()
            torch.cuda.empty_cache()
        log_dist(f"offload_states: moved {moved / 1e9:.2f} GB to {device}")
        return moved

    @torch.no_grad()
    def reload_states(self, non_blocking=False):
        """Inverse of offload_states (reference engine.py:3876)."""
        for container, key, t in list(self._optimizer_state_tensors()):
            if t.is_cuda:
                continue
            dev = torch.empty(t.shape, dtype=t.dtype, device=self.device)
            dev.copy_(t, non_blocking=non_blocking)
            if isinstance(t, torch.nn.Parameter):
                t.data = dev
            else:
                container[key] = dev
        if accel.available():
            accel.synchronize()

    def compile(self, backend="inductor", compile_kwargs=None):
        """torch.compile the wrapped module in place (reference :3820).
        ZeRO-3's .data-swapping hooks are graph breaks by construction, so
        compilation is per-submodule-region under stage 3."""
        self.module = torch.compile(self.module, backend=backend,
                                    **(compile_kwargs or {}))
        self._is_compiled = True
        return self

    @property
    def is_compiled(self) -> bool:
        return getattr(self, "_is_compiled", False)

    # ----------------------------------------------------------- data loader

    def deepspeed_io(self, dataset, batch_size=None, num_workers=0,
                     collate_fn=None, difficulties=None):
        """Build the training dataloader. With
        ``data_efficiency.curriculum_learning`` enabled, samples batch
        indices through the curriculum-aware DeepSpeedDataSampler
        (difficulty = sample length by default, or the DataAnalyzer index
        named by ``metric_path``) — reference engine.deepspeed_io:1831 +
        data_pipeline/data_sampler.py."""
        de = self.config.data_efficiency
        cl = de.curriculum_learning
        if de.enabled and cl.enabled:
            from .data_pipeline.curriculum_scheduler import \
                CurriculumScheduler
            from .data_pipeline.data_sampler import DeepSpeedDataSampler
            if difficulties is None:
                if cl.metric_path:
                    from .data_pipeline.data_analyzer import \
                        load_index_to_metric
                    difficulties = load_index_to_metric(
                        cl.metric_path, cl.difficulty_metric)
                else:
                    difficulties = [
                        len(dataset[i][0]) if isinstance(dataset[i],
                                                         (tuple, list))
                        else len(dataset[i]) for i in range(len(dataset))]
            sched = CurriculumScheduler({
                "curriculum_type": cl.curriculum_type,
                "min_difficulty": cl.min_difficulty,
                "max_difficulty": cl.max_difficulty,
                "schedule_config": dict(cl.schedule_config)})
            self.curriculum_scheduler = sched
            sampler = DeepSpeedDataSampler(
                difficulties, sched,
                batch_size=batch_size or
                self.train_micro_batch_size_per_gpu,
                dp_rank=dist.get_rank(self.dp_group),
                dp_size=self.dp_world_size, seed=de.seed)
            self.curriculum_sampler = sampler
            from torch.utils.data import DataLoader
            post = getattr(self, "_data_post_process_func", None)
            if post is not None:
                from torch.utils.data._utils.collate import default_collate
                base = collate_fn or default_collate

                def collate_fn(items, _base=base, _sched=sched):
                    return post(_base(items),
                                _sched.get_current_difficulty())
            return DataLoader(dataset, batch_sampler=sampler,
                              num_workers=num_workers,
                              collate_fn=collate_fn)
        from .dataloader import build_dataloader
        return build_dataloader(
            dataset,
            batch_size=batch_size or self.train_micro_batch_size_per_gpu,
            world_size=self.dp_world_size,
            rank=dist.get_rank(self.dp_group),
            num_workers=num_workers,
            collate_fn=collate_fn)

    # ----------------------------------------------------------- checkpoints

    def _ckpt_tag(self, tag):
        return tag if tag is not None else f"global_step{self.global_steps}"

    def _model_ckpt_name(self, dirname):
        mp_rank = groups.get_tensor_parallel_rank()
        return os.path.join(dirname, f"mp_rank_{mp_rank:02d}_model_states.pt")

    def _zero_ckpt_name(self, dirname):
        mp_rank = groups.get_tensor_parallel_rank()
        dp_rank = dist.get_rank(self.dp_group)
        return os.path.join(
            dirname, f"zero_pp_rank_{dp_rank}_mp_rank_{mp_rank:02d}_optim_states.pt")

    def save_checkpoint(self, save_dir, tag=None, client_state=None,
                        save_latest=True, exclude_frozen_parameters=False):
        """Reference layout: <dir>/<tag>/mp_rank_XX_model_states.pt +
        zero_pp_rank_R_mp_rank_XX_optim_states.pt + <dir>/latest."""
        tag = self._ckpt_tag(tag)
        ckpt_dir = os.path.join(save_dir, tag)
        os.makedirs(ckpt_dir, exist_ok=True)
        dp_rank = dist.get_rank(self.dp_group)

        if dp_rank == 0:
            module_sd = self.module_state_dict(
                exclude_frozen_parameters=exclude_frozen_parameters)
            state = {
                "module": module_sd,
                "buffer_names": [n for n, _ in self.module.named_buffers()],
                "lr_scheduler": (self.lr_scheduler.state_dict()
                                 if self.lr_scheduler is not None and
                                 hasattr(self.lr_scheduler, "state_dict") else None),
                "global_steps": self.global_steps,
                "global_samples": self.global_samples,
                "skipped_steps": self.skipped_steps,
                "dp_world_size": self.dp_world_size,
                "random_ltd": (self.random_ltd_scheduler.state_dict()
                               if self.random_ltd_scheduler is not None
                               else None),
                "ds_config": self.config.raw,
                "client_state": client_state or {},
            }
            self.checkpoint_engine.save(state, self._model_ckpt_name(ckpt_dir))

        if self.has_moe_layers:
            self._save_moe_checkpoint(ckpt_dir)

        if self.optimizer is not None and hasattr(self.optimizer, "state_dict") and \
                not isinstance(self.optimizer, DummyOptim):
            opt_state = {"optimizer_state_dict": self.optimizer.state_dict()}
            self.checkpoint_engine.save(opt_state, self._zero_ckpt_name(ckpt_dir))

        dist.barrier()
        ok = self.checkpoint_engine.commit(tag)
        if save_latest and self.global_rank == 0 and ok:
            with open(os.path.join(save_dir, "latest"), "w") as f:
                f.write(tag)
        dist.barrier()
        return True

    def _expert_state(self):
        """This rank's expert parameters/buffers (tagged allreduce=False)."""
        out = {}
        for n, p in self.module.named_parameters():
            if getattr(p, "allreduce", True) is False:
                out[n] = p.detach().cpu()
        return out

    def _moe_ckpt_name(self, ckpt_dir, ep_rank):
        return os.path.join(ckpt_dir,
                            f"expert_ep_rank_{ep_rank}_model_states.pt")

    def _save_moe_checkpoint(self, ckpt_dir):
        """With EP > 1 every EP rank owns DIFFERENT experts, so the dense
        mp_rank_00 file (written by dp-rank 0 only) cannot carry them:
        each expert shard is saved once by its expert-data-parallel rank 0
        (reference engine._save_moe_checkpoint:3319)."""
        from ..moe.layer import MoE
        names = {m.expert_group_name for m in self.module.modules()
                 if isinstance(m, MoE)}
        if not names:
            return
        name = next(iter(names))
        edp = groups.get_expert_data_parallel_group(name)
        if dist.get_rank(edp) == 0:
            ep_rank = groups.get_expert_parallel_rank(name)
            self.checkpoint_engine.save(self._expert_state(),
                                        self._moe_ckpt_name(ckpt_dir,
                                                            ep_rank))

    def _load_moe_checkpoint(self, ckpt_dir):
        from ..moe.layer import MoE
        names = {m.expert_group_name for m in self.module.modules()
                 if isinstance(m, MoE)}
        if not names:
            return
        name = next(iter(names))
        ep_rank = groups.get_expert_parallel_rank(name)
        path = self._moe_ckpt_name(ckpt_dir, ep_rank)
        if not os.path.exists(path):
            logger.warning(f"no expert checkpoint for ep_rank {ep_rank}")
            return
        expert_sd = self.checkpoint_engine.load(path, map_location="cpu")
        missing = self.module.load_state_dict(expert_sd, strict=False)
        del missing  # only experts in this file by construction

    def module_state_dict(self, exclude_frozen_parameters=False):
        if self.zero_stage == 3 and hasattr(self.optimizer, "full_state_dict"):
            sd = self.optimizer.full_state_dict()
        else:
            sd = self.module.state_dict()
        if exclude_frozen_parameters:
            frozen = {n for n, p in self.module.named_parameters()
                      if not p.requires_grad}
            sd = {k: v for k, v in sd.items() if k not in frozen}
        return sd

    def load_checkpoint(self, load_dir, tag=None, load_module_strict=True,
                        load_optimizer_states=True, load_lr_scheduler_states=True,
                        load_module_only=False, load_universal=False):
        if tag is None:
            latest = os.path.join(load_dir, "latest")
            if not os.path.exists(latest):
                logger.warning(f"no 'latest' file in {load_dir}")
                return None, {}
            with open(latest) as f:
                tag = f.read().strip()
        ckpt_dir = os.path.join(load_dir, tag)
        model_path = self._model_ckpt_name(ckpt_dir)
        state = self.checkpoint_engine.load(model_path, map_location="cpu")

        if self.zero_stage == 3 and hasattr(self.optimizer, "load_full_state_dict"):
            self.optimizer.load_full_state_dict(state["module"],
                                                strict=load_module_strict)
        else:
            self.module.load_state_dict(
                state["module"],
                strict=load_module_strict and not self.has_moe_layers)
        if self.has_moe_layers:
            self._load_moe_checkpoint(ckpt_dir)
        self.global_steps = state.get("global_steps", 0)
        self.global_samples = state.get("global_samples", 0)
        self.skipped_steps = state.get("skipped_steps", 0)
        if self.random_ltd_scheduler is not None and \
                state.get("random_ltd") is not None:
            self.random_ltd_scheduler.load_state_dict(state["random_ltd"])

        if load_universal:
            # elastic load: per-param fp32 state sliced at THIS world size
            from ..checkpoint.universal import load_universal as _load_usd
            usd = _load_usd(os.path.join(load_dir, f"{tag}_universal"))
            assert hasattr(self.optimizer, "load_universal_state_dict"), \
                "universal load requires a ZeRO optimizer"
            self.optimizer.load_universal_state_dict(self.module, usd)
            return ckpt_dir, state.get("client_state", {})

        if not load_module_only:
            if load_lr_scheduler_states and self.lr_scheduler is not None and \
                    state.get("lr_scheduler") is not None:
                self.lr_scheduler.load_state_dict(state["lr_scheduler"])
            zero_path = self._zero_ckpt_name(ckpt_dir)
            if os.path.exists(zero_path) and self.optimizer is not None and \
                    hasattr(self.optimizer, "load_state_dict"):
                opt_state = self.checkpoint_engine.load(zero_path,
                                                        map_location="cpu")
                if hasattr(self.optimizer, "loss_scaler"):
                    self.optimizer.load_state_dict(
                        opt_state["optimizer_state_dict"],
                        load_optimizer_states=load_optimizer_states)
                else:
                    self.optimizer.load_state_dict(opt_state["optimizer_state_dict"])
        # ZeRO-1/2 load re-broadcasts masters into the 16-bit params; for
        # stage 0 re-broadcast from rank 0 for determinism.
        if not hasattr(self.optimizer, "loss_scaler"):
            self._broadcast_model()
        return ckpt_dir, state.get("client_state", {})

    def save_16bit_model(self, save_dir, save_filename="pytorch_model.bin",
                         exclude_frozen_parameters=False):
        if self.zero_stage == 3 and \
                not self.config.zero.stage3_gather_16bit_weights_on_model_save:
            logger.warning(
                "save_16bit_model skipped: ZeRO-3 params are partitioned and "
                "stage3_gather_16bit_weights_on_model_save is false "
                "(reference engine.save_16bit_model behavior)")
            return False
        sd = self.module_state_dict(exclude_frozen_parameters)
        if self.global_rank == 0:
            os.makedirs(save_dir, exist_ok=True)
            torch.save(sd, os.path.join(save_dir, save_filename))
        dist.barrier()
        return True
