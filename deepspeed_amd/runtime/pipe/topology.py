"""Pipeline process topology (reference: deepspeed/runtime/pipe/topology.py
PipeDataParallelTopology :232, PipeModelDataParallelTopology :244 /
PipelineParallelGrid :251).

Rank layout is pipe-major with tensor-parallel innermost:
``rank = (stage * dp_size + dp_rank) * tp_size + tp_rank``. On one MI355X
node every pair of GPUs has a direct xGMI link, so there is no locality
penalty for any layout; TP-innermost keeps the latency-sensitive TP
all-reduces between adjacent (same-node) ranks and each stage's DP group
contiguous-strided, which is the shape RCCL's ring likes for the
per-stage reduce-scatter.
"""

from typing import List

from ... import comm as dist
from ...parallel import groups as pgroups


class PipelineParallelGrid:
    """Builds and owns the pipe / data / tensor process groups.

    Exposes the mpu interface the engine consumes
    (get_data_parallel_group / get_model_parallel_group) so the ZeRO
    optimizer partitions over THIS (stage, tp) cell's data-parallel
    replicas only, and norms reduce over the full non-DP (pipe x tensor)
    complement.
    """

    def __init__(self, num_stages: int, world_size: int = None,
                 tp_size: int = 1):
        world_size = world_size or dist.get_world_size()
        assert world_size % (num_stages * tp_size) == 0, \
            f"world {world_size} not divisible by {num_stages} stages " \
            f"x tp {tp_size}"
        self.pipe_parallel_size = num_stages
        self.tensor_parallel_size = tp_size
        self.data_parallel_size = world_size // (num_stages * tp_size)
        self.world_size = world_size
        self.global_rank = dist.get_rank()
        cell = self.data_parallel_size * tp_size   # ranks per stage
        self.stage_id = self.global_rank // cell
        self.data_parallel_id = (self.global_rank % cell) // tp_size
        self.tensor_parallel_id = self.global_rank % tp_size

        self.dp_group = None
        self.pp_group = None
        self.tp_group = None
        self.mp_group = None     # pipe x tensor complement of my dp group
        self.dp_groups: List = []
        self.pp_groups: List = []

        def rank_of(stage, dp, tp):
            return (stage * self.data_parallel_size + dp) * tp_size + tp

        for stage in range(num_stages):
            for tp in range(tp_size):
                ranks = [rank_of(stage, d, tp)
                         for d in range(self.data_parallel_size)]
                g = dist.new_group(ranks)
                self.dp_groups.append((ranks, g))
                if self.global_rank in ranks:
                    self.dp_group = g
        for d in range(self.data_parallel_size):
            for tp in range(tp_size):
                ranks = [rank_of(s, d, tp) for s in range(num_stages)]
                g = dist.new_group(ranks)
                self.pp_groups.append((ranks, g))
                if self.global_rank in ranks:
                    self.pp_group = g
                    self.pp_ranks = ranks
        if tp_size > 1:
            for stage in range(num_stages):
                for d in range(self.data_parallel_size):
                    ranks = [rank_of(stage, d, t) for t in range(tp_size)]
                    g = dist.new_group(ranks)
                    if self.global_rank in ranks:
                        self.tp_group = g
            # register with the global TP registry so Column/RowParallel
            # layers inside pipeline stages (and checkpoint mp_rank naming)
            # resolve THIS grid's tensor groups
            pgroups.set_tensor_parallel_group(self.tp_group, tp_size,
                                              self.tensor_parallel_id)
        if tp_size > 1 or num_stages > 1:
            for d in range(self.data_parallel_size):
                ranks = sorted(rank_of(s, d, t) for s in range(num_stages)
                               for t in range(tp_size))
                g = dist.new_group(ranks)
                if self.global_rank in ranks:
                    self.mp_group = g

    # ---- mpu interface (engine / ZeRO consume these) ----
    def get_data_parallel_group(self):
        return self.dp_group

    def get_data_parallel_rank(self):
        return self.data_parallel_id

    def get_data_parallel_world_size(self):
        return self.data_parallel_size

    def get_model_parallel_group(self):
        # everything that is not data-parallel: norms computed on one
        # (stage, tp) shard must reduce over pipe AND tensor ranks
        return self.mp_group if self.mp_group is not None else self.pp_group

    def get_model_parallel_world_size(self):
        return self.pipe_parallel_size * self.tensor_parallel_size

    def get_model_parallel_rank(self):
        return self.stage_id * self.tensor_parallel_size + \
            self.tensor_parallel_id

    def get_tensor_parallel_group(self):
        return self.tp_group

    def get_tensor_parallel_rank(self):
        return self.tensor_parallel_id

    def get_tensor_parallel_world_size(self):
        return self.tensor_parallel_size

    def get_pipe_parallel_group(self):
        return self.pp_group

    def get_pipe_parallel_rank(self):
        return self.stage_id

    def get_pipe_parallel_world_size(self):
        return self.pipe_parallel_size

    # ---- p2p neighbours ----
    def stage_to_global(self, stage: int) -> int:
        return (stage * self.data_parallel_size + self.data_parallel_id) * \
            self.tensor_parallel_size + self.tensor_parallel_id

    @property
    def prev_stage_rank(self):
        return self.stage_to_global(self.stage_id - 1)

    @property
    def next_stage_rank(self):
        return self.stage_to_global(self.stage_id + 1)

    def is_first_stage(self) -> bool:
        return self.stage_id == 0

    def is_last_stage(self) -> bool:
        return self.stage_id == self.pipe_parallel_size - 1
