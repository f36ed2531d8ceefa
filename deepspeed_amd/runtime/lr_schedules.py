"""LR schedules: LRRangeTest, OneCycle, WarmupLR, WarmupDecayLR, WarmupCosineLR.

Capability parity with the reference's ``deepspeed/runtime/lr_schedules.py``
(:18-22). Each schedule steps per optimizer step and supports
state_dict/load_state_dict for checkpointing.
"""

import math
from typing import List

VALID_SCHEDULES = ["LRRangeTest", "OneCycle", "WarmupLR", "WarmupDecayLR",
                   "WarmupCosineLR"]


def _to_list(x, n):
    if isinstance(x, (list, tuple)):
        return list(x)
    return [x] * n


class _BaseSchedule:
    def __init__(self, optimizer, last_batch_iteration=-1):
        self.optimizer = optimizer
        self.last_batch_iteration = last_batch_iteration
        if last_batch_iteration == -1:
            self.step(0)

    def get_lr(self) -> List[float]:
        raise NotImplementedError

    def get_last_lr(self):
        return self._last_lr

    def step(self, last_batch_iteration=None):
        if last_batch_iteration is None:
            last_batch_iteration = self.last_batch_iteration + 1
        self.last_batch_iteration = last_batch_iteration
        lrs = self.get_lr()
        for group, lr in zip(self.optimizer.param_groups, lrs):
            group["lr"] = lr
        self._last_lr = lrs

    def state_dict(self):
        return {"last_batch_iteration": self.last_batch_iteration}

    def load_state_dict(self, sd):
        self.last_batch_iteration = sd["last_batch_iteration"]
        self.step(self.last_batch_iteration)


class WarmupLR(_BaseSchedule):
    """Linear warmup from warmup_min_lr to warmup_max_lr, then constant."""

    def __init__(self, optimizer, warmup_min_lr=0.0, warmup_max_lr=0.001,
                 warmup_num_steps=1000, warmup_type="log", last_batch_iteration=-1):
        n = len(optimizer.param_groups)
        self.min_lrs = _to_list(warmup_min_lr, n)
        self.max_lrs = _to_list(warmup_max_lr, n)
        self.warmup_num_steps = max(2, warmup_num_steps)
        self.warmup_type = warmup_type
        self.inverse_log_warm_up = 1.0 / math.log(self.warmup_num_steps)
        super().__init__(optimizer, last_batch_iteration)

    def _gamma(self):
        step = self.last_batch_iteration + 1
        if step < self.warmup_num_steps:
            if self.warmup_type == "log":
                return self.inverse_log_warm_up * math.log(max(step, 1))
            return min(1.0, step / self.warmup_num_steps)
        return 1.0

    def get_lr(self):
        g = self._gamma()
        return [lo + g * (hi - lo) for lo, hi in zip(self.min_lrs, self.max_lrs)]


class WarmupDecayLR(WarmupLR):
    """Warmup then linear decay to 0 at total_num_steps."""

    def __init__(self, optimizer, total_num_steps, warmup_min_lr=0.0,
                 warmup_max_lr=0.001, warmup_num_steps=1000, warmup_type="log",
                 last_batch_iteration=-1):
        self.total_num_steps = total_num_steps
        super().__init__(optimizer, warmup_min_lr, warmup_max_lr, warmup_num_steps,
                         warmup_type, last_batch_iteration)

    def _gamma(self):
        step = self.last_batch_iteration + 1
        if step < self.warmup_num_steps:
            return super()._gamma()
        return max(0.0, (self.total_num_steps - step) /
                   max(1, self.total_num_steps - self.warmup_num_steps))


class WarmupCosineLR(_BaseSchedule):
    """Linear warmup then cosine decay to cos_min_ratio of peak."""

    def __init__(self, optimizer, total_num_steps, warmup_min_ratio=0.0,
                 warmup_num_steps=1000, cos_min_ratio=0.0001,
                 last_batch_iteration=-1):
        self.total_num_steps = total_num_steps
        self.warmup_min_ratio = warmup_min_ratio
        self.warmup_num_steps = max(1, warmup_num_steps)
        self.cos_min_ratio = cos_min_ratio
        self.org_lrs = [g["lr"] for g in optimizer.param_groups]
        super().__init__(optimizer, last_batch_iteration)

    def get_lr(self):
        step = self.last_batch_iteration + 1
        if step < self.warmup_num_steps:
            ratio = self.warmup_min_ratio + (1 - self.warmup_min_ratio) * (
                step / self.warmup_num_steps)
        else:
            progress = min(1.0, (step - self.warmup_num_steps) /
                           max(1, self.total_num_steps - self.warmup_num_steps))
            cos = 0.5 * (1 + math.cos(math.pi * progress))
            ratio = self.cos_min_ratio + (1 - self.cos_min_ratio) * cos
        return [lr * ratio for lr in self.org_lrs]


class LRRangeTest(_BaseSchedule):
    """LR range test: ramp LR (linearly or continuously) for tuning."""

    def __init__(self, optimizer, lr_range_test_min_lr=1e-3,
                 lr_range_test_step_size=2000, lr_range_test_step_rate=1.0,
                 lr_range_test_staircase=False, last_batch_iteration=-1):
        n = len(optimizer.param_groups)
        self.min_lrs = _to_list(lr_range_test_min_lr, n)
        self.step_size = lr_range_test_step_size
        self.step_rate = lr_range_test_step_rate
        self.staircase = lr_range_test_staircase
        super().__init__(optimizer, last_batch_iteration)

    def get_lr(self):
        step = self.last_batch_iteration + 1
        if self.staircase:
            interval = math.floor(step / self.step_size)
        else:
            interval = step / self.step_size
        scale = 1.0 + self.step_rate * interval
        return [lr * scale for lr in self.min_lrs]


class OneCycle(_BaseSchedule):
    """1-cycle LR (and optional momentum) policy."""

    def __init__(self, optimizer, cycle_min_lr, cycle_max_lr,
                 decay_lr_rate=0.0, cycle_first_step_size=2000,
                 cycle_second_step_size=None, cycle_first_stair_count=0,
                 cycle_second_stair_count=None, decay_step_size=0,
                 cycle_momentum=True, cycle_min_mom=0.8, cycle_max_mom=0.9,
                 decay_mom_rate=0.0, last_batch_iteration=-1):
        n = len(optimizer.param_groups)
        self.min_lrs = _to_list(cycle_min_lr, n)
        self.max_lrs = _to_list(cycle_max_lr, n)
        self.decay_lr_rate = decay_lr_rate
        self.first_size = cycle_first_step_size
        self.second_size = (cycle_second_step_size if cycle_second_step_size is not None
                            else cycle_first_step_size)
        self.decay_step_size = decay_step_size
        self.cycle_momentum = cycle_momentum and "betas" in optimizer.defaults
        self.min_mom, self.max_mom = cycle_min_mom, cycle_max_mom
        self.decay_mom_rate = decay_mom_rate
        super().__init__(optimizer, last_batch_iteration)

    def _phase(self):
        step = self.last_batch_iteration + 1
        total = self.first_size + self.second_size
        if step <= self.first_size:
            return step / self.first_size, "up"
        if step <= total:
            return (step - self.first_size) / self.second_size, "down"
        return step - total, "decay"

    def get_lr(self):
        x, phase = self._phase()
        lrs = []
        for lo, hi in zip(self.min_lrs, self.max_lrs):
            if phase == "up":
                lrs.append(lo + x * (hi - lo))
            elif phase == "down":
                lrs.append(hi - x * (hi - lo))
            else:
                if self.decay_step_size > 0:
                    decay = (1 + self.decay_lr_rate) ** -(x / self.decay_step_size)
                else:
                    decay = 1.0 / (1 + self.decay_lr_rate * x)
                lrs.append(lo * decay)
        return lrs

    def step(self, last_batch_iteration=None):
        super().step(last_batch_iteration)
        if self.cycle_momentum:
            x, phase = self._phase()
            if phase == "up":
                mom = self.max_mom - x * (self.max_mom - self.min_mom)
            elif phase == "down":
                mom = self.min_mom + x * (self.max_mom - self.min_mom)
            else:
                mom = self.max_mom
            for group in self.optimizer.param_groups:
                if "betas" in group:
                    group["betas"] = (mom, group["betas"][1])


SCHEDULE_CLASSES = {
    "WarmupLR": WarmupLR,
    "WarmupDecayLR": WarmupDecayLR,
    "WarmupCosineLR": WarmupCosineLR,
    "LRRangeTest": LRRangeTest,
    "OneCycle": OneCycle,
}


def get_scheduler(name: str, optimizer, params: dict):
    if name not in SCHEDULE_CLASSES:
        raise ValueError(f"unknown scheduler {name}; valid: {VALID_SCHEDULES}")
    return SCHEDULE_CLASSES[name](optimizer, **params)
