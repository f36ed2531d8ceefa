"""Curriculum-aware distributed data sampler (reference:
deepspeed/runtime/data_pipeline/data_sampler.py DeepSpeedDataSampler :36).

Each sample carries a difficulty value (e.g. sequence length); at global
step t only samples with difficulty <= scheduler.current_difficulty are
eligible. Eligible indices are deterministically shuffled per epoch and
strided across the DP group."""

from typing import Optional, Sequence

import torch

from .curriculum_scheduler import CurriculumScheduler


class DeepSpeedDataSampler:
    def __init__(self, difficulties: Sequence[float],
                 curriculum: Optional[CurriculumScheduler],
                 batch_size: int, dp_rank: int = 0, dp_size: int = 1,
                 seed: int = 1234, drop_last: bool = True):
        self.difficulties = torch.as_tensor(list(difficulties),
                                            dtype=torch.float64)
        self.curriculum = curriculum
        self.batch_size = batch_size
        self.dp_rank = dp_rank
        self.dp_size = dp_size
        self.seed = seed
        self.drop_last = drop_last
        self.epoch = 0
        self.global_step = 0

    def set_epoch(self, epoch: int):
        self.epoch = epoch

    def state_dict(self):
        return {"epoch": self.epoch, "global_step": self.global_step,
                "curriculum": self.curriculum.state_dict()
                if self.curriculum else None}

    def load_state_dict(self, sd):
        self.epoch = sd["epoch"]
        self.global_step = sd["global_step"]
        if self.curriculum and sd.get("curriculum"):
            self.curriculum.load_state_dict(sd["curriculum"])

    def _eligible(self):
        if self.curriculum is None:
            return torch.arange(len(self.difficulties))
        cur = self.curriculum.update_difficulty(self.global_step)
        return (self.difficulties <= cur).nonzero(as_tuple=True)[0]

    def __iter__(self):
        g = torch.Generator().manual_seed(self.seed + self.epoch)
        order = torch.randperm(len(self.difficulties), generator=g)
        i = 0
        gbs = self.batch_size * self.dp_size
        while i < len(order):
            eligible = set(self._eligible().tolist())
            batch = []
            j = i
            while j < len(order) and len(batch) < gbs:
                idx = int(order[j])
                if idx in eligible:
                    batch.append(idx)
                j += 1
            i = j
            if len(batch) < gbs:
                if self.drop_last or not batch:
                    break
            mine = batch[self.dp_rank::self.dp_size][:self.batch_size]
            self.global_step += 1
            yield mine

    def __len__(self):
        return len(self.difficulties) // (self.batch_size * self.dp_size)
