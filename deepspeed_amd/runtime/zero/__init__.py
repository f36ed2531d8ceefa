"""ZeRO optimizers: stage 1/2 (flat-bucket grad/state partitioning) and
stage 3 (module-unit parameter partitioning) + the Init/GatheredParameters
public API."""

from .partition import GatheredParameters, Init
from .stage12 import ZeroStage12Optimizer
from .stage3 import ZeroStage3Optimizer

__all__ = ["ZeroStage12Optimizer", "ZeroStage3Optimizer", "Init",
           "GatheredParameters"]


def estimate_zero3_model_states_mem_needs(total_params, num_gpus_per_node=8,
                                          num_nodes=1, cpu_offload=False,
                                          additional_buffer_factor=1.5):
    """Per-GPU memory needs for ZeRO-3 (reference stage3.py
    estimate_zero3_model_states_mem_needs...): params bf16 (sharded) +
    grads bf16 (sharded) + fp32 master/m/v (sharded, or host when
    offloaded). Returns (gpu_bytes, cpu_bytes)."""
    world = num_gpus_per_node * num_nodes
    shard = total_params / world
    live_params = 2 * shard          # bf16 shard (gathers are transient)
    grads = 2 * shard
    states = 12 * shard              # fp32 master + exp_avg + exp_avg_sq
    if cpu_offload:
        gpu = (live_params + grads) * additional_buffer_factor
        cpu = states * world / num_nodes * additional_buffer_factor
    else:
        gpu = (live_params + grads + states) * additional_buffer_factor
        cpu = 0.0
    return int(gpu), int(cpu)


def estimate_zero2_model_states_mem_needs(total_params, num_gpus_per_node=8,
                                          num_nodes=1, cpu_offload=False,
                                          additional_buffer_factor=1.5):
    """ZeRO-2: full bf16 params + full bf16 grads per GPU, optimizer states
    sharded (reference stage_1_and_2.py estimator)."""
    world = num_gpus_per_node * num_nodes
    params = 2 * total_params
    grads = 2 * total_params
    states = 12 * total_params / world
    if cpu_offload:
        gpu = (params + grads) * additional_buffer_factor
        cpu = states * world / num_nodes * additional_buffer_factor
    else:
        gpu = (params + grads + states) * additional_buffer_factor
        cpu = 0.0
    return int(gpu), int(cpu)
