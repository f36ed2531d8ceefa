"""Process-group topology: DP / TP / SP / EP group construction.

Capability parity with the reference's ``deepspeed/utils/groups.py``
(expert / expert-data / model-parallel group factories) and the mesh-based
DP x SP groups. One MI355X node is a full xGMI mesh (7 p2p links per GPU),
so any partitioning of the 8 ranks has equal link bandwidth — group layout
is chosen for collective *shape* (a2a-heavy groups like EP/SP benefit from
the full mesh), not for physical locality.
"""

from typing import Dict, Optional

import torch

from .. import comm as dist
from ..utils.logging import log_dist

_WORLD_GROUP = None
_DATA_PARALLEL_GROUP = None
_MODEL_PARALLEL_GROUP = None
_TENSOR_PARALLEL_GROUP = None
_SEQUENCE_PARALLEL_GROUP = None
_SEQUENCE_DATA_PARALLEL_GROUP = None
_EXPERT_PARALLEL_GROUPS: Dict[str, object] = {}
_EXPERT_DATA_PARALLEL_GROUPS: Dict[str, object] = {}
_mpu = None


def reset_groups():
    """Testing hook: forget all cached groups (does not destroy them)."""
    global _WORLD_GROUP, _DATA_PARALLEL_GROUP, _MODEL_PARALLEL_GROUP
    global _TENSOR_PARALLEL_GROUP, _SEQUENCE_PARALLEL_GROUP
    global _SEQUENCE_DATA_PARALLEL_GROUP, _mpu
    _WORLD_GROUP = None
    _DATA_PARALLEL_GROUP = None
    _MODEL_PARALLEL_GROUP = None
    _TENSOR_PARALLEL_GROUP = None
    _SEQUENCE_PARALLEL_GROUP = None
    _SEQUENCE_DATA_PARALLEL_GROUP = None
    _EXPERT_PARALLEL_GROUPS.clear()
    _EXPERT_DATA_PARALLEL_GROUPS.clear()
    _mpu = None


def set_mpu(mpu):
    """Install an external model-parallel unit (Megatron-style interface)."""
    global _mpu
    _mpu = mpu


def _ensure_world_group():
    global _WORLD_GROUP
    if _WORLD_GROUP is None and dist.is_initialized():
        _WORLD_GROUP = torch.distributed.group.WORLD
    return _WORLD_GROUP


def get_data_parallel_group():
    if _mpu is not None:
        return _mpu.get_data_parallel_group()
    if _DATA_PARALLEL_GROUP is not None:
        return _DATA_PARALLEL_GROUP
    if _SEQUENCE_PARALLEL_GROUP is not None:
        # Ulysses: the model is replicated over SP and each SP rank holds a
        # different sequence chunk, so ZeRO must shard AND average grads
        # over the full DPxSP mesh — the average over world equals
        # (1/dp)sum_dp (1/sp)sum_sp d(local mean loss), i.e. the exact
        # gradient of the global token-mean loss with NO extra scaling.
        return _ensure_world_group()
    return _ensure_world_group()


def get_data_parallel_world_size() -> int:
    return dist.get_world_size(get_data_parallel_group())


def get_data_parallel_rank() -> int:
    return dist.get_rank(get_data_parallel_group())


def get_model_parallel_group():
    if _mpu is not None:
        return _mpu.get_model_parallel_group()
    return _MODEL_PARALLEL_GROUP


def get_model_parallel_world_size() -> int:
    g = get_model_parallel_group()
    return dist.get_world_size(g) if g is not None else 1


def get_tensor_parallel_group():
    if _mpu is not None and hasattr(_mpu, "get_tensor_model_parallel_group"):
        return _mpu.get_tensor_model_parallel_group()
    return _TENSOR_PARALLEL_GROUP


def get_tensor_parallel_world_size() -> int:
    g = get_tensor_parallel_group()
    return dist.get_world_size(g) if g is not None else 1


def get_tensor_parallel_rank() -> int:
    g = get_tensor_parallel_group()
    return dist.get_rank(g) if g is not None else 0


def get_sequence_parallel_group():
    return _SEQUENCE_PARALLEL_GROUP


def get_sequence_parallel_world_size() -> int:
    g = _SEQUENCE_PARALLEL_GROUP
    return dist.get_world_size(g) if g is not None else 1


def get_sequence_parallel_rank() -> int:
    g = _SEQUENCE_PARALLEL_GROUP
    return dist.get_rank(g) if g is not None else 0


def set_tensor_parallel_group(group, world_size=None, rank=None):
    """Register an externally-built TP group (e.g. the pipeline grid's
    per-(stage, dp) tensor groups) so Column/RowParallel layers and
    mp_rank checkpoint naming resolve it. world_size/rank are accepted
    for symmetry but derived from the group when queried."""
    global _TENSOR_PARALLEL_GROUP, _MODEL_PARALLEL_GROUP
    _TENSOR_PARALLEL_GROUP = group
    _MODEL_PARALLEL_GROUP = group


def initialize_tensor_parallel(tp_size: int):
    """Create TP groups: ranks [i*tp, (i+1)*tp) form one TP group; DP group is
    the strided complement."""
    global _TENSOR_PARALLEL_GROUP, _DATA_PARALLEL_GROUP, _MODEL_PARALLEL_GROUP
    world = dist.get_world_size()
    assert world % tp_size == 0, f"world {world} not divisible by tp {tp_size}"
    rank = dist.get_rank()
    for start in range(0, world, tp_size):
        ranks = list(range(start, start + tp_size))
        g = dist.new_group(ranks)
        if rank in ranks:
            _TENSOR_PARALLEL_GROUP = g
            _MODEL_PARALLEL_GROUP = g
    dp_size = world // tp_size
    for tp_rank in range(tp_size):
        ranks = list(range(tp_rank, world, tp_size))
        g = dist.new_group(ranks)
        if rank in ranks:
            _DATA_PARALLEL_GROUP = g
    log_dist(f"TP groups initialized: tp={tp_size} dp={dp_size}")
    return _TENSOR_PARALLEL_GROUP


def initialize_sequence_parallel(sp_size: int):
    """Ulysses DPxSP mesh: contiguous ranks form an SP group; the strided
    complement forms the (sequence-)data-parallel group used for ZeRO."""
    global _SEQUENCE_PARALLEL_GROUP, _SEQUENCE_DATA_PARALLEL_GROUP
    world = dist.get_world_size()
    assert world % sp_size == 0, f"world {world} not divisible by sp {sp_size}"
    rank = dist.get_rank()
    for start in range(0, world, sp_size):
        ranks = list(range(start, start + sp_size))
        g = dist.new_group(ranks)
        if rank in ranks:
            _SEQUENCE_PARALLEL_GROUP = g
    for sp_rank in range(sp_size):
        ranks = list(range(sp_rank, world, sp_size))
        g = dist.new_group(ranks)
        if rank in ranks:
            _SEQUENCE_DATA_PARALLEL_GROUP = g
    log_dist(f"SP groups initialized: sp={sp_size} dp={world // sp_size}")
    return _SEQUENCE_PARALLEL_GROUP


def initialize_expert_parallel(ep_size: int, name: Optional[str] = None):
    """Create EP groups (contiguous) + expert-DP groups (strided complement).

    Mirrors ``_create_expert_and_data_parallel`` (reference groups.py:236):
    expert parallelism slices the DP world; each expert's parameters are
    data-parallel across the ranks that hold the same expert slice.
    """
    name = name or f"ep_size_{ep_size}"
    if name in _EXPERT_PARALLEL_GROUPS:
        return _EXPERT_PARALLEL_GROUPS[name]
    world = dist.get_world_size()
    ep_size = min(ep_size, world)
    assert world % ep_size == 0, f"world {world} not divisible by ep {ep_size}"
    rank = dist.get_rank()
    for start in range(0, world, ep_size):
        ranks = list(range(start, start + ep_size))
        g = dist.new_group(ranks)
        if rank in ranks:
            _EXPERT_PARALLEL_GROUPS[name] = g
    for ep_rank in range(ep_size):
        ranks = list(range(ep_rank, world, ep_size))
        g = dist.new_group(ranks)
        if rank in ranks:
            _EXPERT_DATA_PARALLEL_GROUPS[name] = g
    log_dist(f"EP groups initialized: ep={ep_size} name={name}")
    return _EXPERT_PARALLEL_GROUPS[name]


def get_expert_parallel_group(name: str):
    return _EXPERT_PARALLEL_GROUPS[name]


def get_expert_data_parallel_group(name: str):
    return _EXPERT_DATA_PARALLEL_GROUPS[name]


def get_expert_parallel_world_size(name: str) -> int:
    return dist.get_world_size(get_expert_parallel_group(name))


def get_expert_parallel_rank(name: str) -> int:
    return dist.get_rank(get_expert_parallel_group(name))


def expert_parallel_group_names():
    return list(_EXPERT_PARALLEL_GROUPS.keys())
