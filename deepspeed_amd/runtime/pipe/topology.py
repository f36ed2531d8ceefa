"""Pipeline process topology (reference: deepspeed/runtime/pipe/topology.py
PipeDataParallelTopology :232 / PipelineParallelGrid :251).

Rank layout is pipe-major: ``rank = stage * dp_size + dp_rank``. On one
MI355X node every pair of GPUs has a direct xGMI link, so there is no
locality penalty for any layout; pipe-major keeps each stage's DP group
contiguous, which is the shape RCCL's ring likes for the per-stage
reduce-scatter.
"""

from typing import List

from ... import comm as dist


class PipelineParallelGrid:
    """Builds and owns the pipe / data process groups.

    Exposes the mpu interface the engine consumes
    (get_data_parallel_group / get_model_parallel_group) so the ZeRO
    optimizer partitions over THIS stage's data-parallel replicas only.
    """

    def __init__(self, num_stages: int, world_size: int = None):
        world_size = world_size or dist.get_world_size()
        assert world_size % num_stages == 0, \
            f"world {world_size} not divisible by {num_stages} stages"
        self.pipe_parallel_size = num_stages
        self.data_parallel_size = world_size // num_stages
        self.world_size = world_size
        self.global_rank = dist.get_rank()
        self.stage_id = self.global_rank // self.data_parallel_size
        self.data_parallel_id = self.global_rank % self.data_parallel_size

        self.dp_group = None
        self.pp_group = None
        self.dp_groups: List = []
        self.pp_groups: List = []
        for stage in range(num_stages):
            ranks = [stage * self.data_parallel_size + d
                     for d in range(self.data_parallel_size)]
            g = dist.new_group(ranks)
            self.dp_groups.append((ranks, g))
            if self.global_rank in ranks:
                self.dp_group = g
        for d in range(self.data_parallel_size):
            ranks = [s * self.data_parallel_size + d for s in range(num_stages)]
            g = dist.new_group(ranks)
            self.pp_groups.append((ranks, g))
            if self.global_rank in ranks:
                self.pp_group = g
                self.pp_ranks = ranks

    # ---- mpu interface (engine / ZeRO consume these) ----
    def get_data_parallel_group(self):
        return self.dp_group

    def get_data_parallel_rank(self):
        return self.data_parallel_id

    def get_data_parallel_world_size(self):
        return self.data_parallel_size

    def get_model_parallel_group(self):
        # no tensor parallelism inside the pipe grid; norms reduce over pipe
        return self.pp_group

    def get_model_parallel_world_size(self):
        return 1

    def get_model_parallel_rank(self):
        return 0

    def get_pipe_parallel_group(self):
        return self.pp_group

    def get_pipe_parallel_rank(self):
        return self.stage_id

    def get_pipe_parallel_world_size(self):
        return self.pipe_parallel_size

    # ---- p2p neighbours ----
    def stage_to_global(self, stage: int) -> int:
        return stage * self.data_parallel_size + self.data_parallel_id

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
