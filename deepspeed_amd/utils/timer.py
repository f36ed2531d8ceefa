"""Wall-clock + throughput timers.

Capability parity with the reference's ``deepspeed/utils/timer.py``
(SynchronizedWallClockTimer / ThroughputTimer), using HIP events through
``torch.cuda`` when a GPU is present.
"""

import time

from .. import accel
from .logging import log_dist

FORWARD_MICRO_TIMER = "fwd_microstep"
FORWARD_GLOBAL_TIMER = "fwd"
BACKWARD_MICRO_TIMER = "bwd_microstep"
BACKWARD_GLOBAL_TIMER = "bwd"
STEP_MICRO_TIMER = "step_microstep"
STEP_GLOBAL_TIMER = "step"


class _Timer:
    def __init__(self, name):
        self.name = name
        self.started = False
        self.elapsed_s = 0.0
        self._start = 0.0
        self.count = 0

    def start(self):
        if self.started:
            return
        accel.synchronize()
        self._start = time.perf_counter()
        self.started = True

    def stop(self, record_count=True):
        if not self.started:
            return
        accel.synchronize()
        self.elapsed_s += time.perf_counter() - self._start
        if record_count:
            self.count += 1
        self.started = False

    def reset(self):
        self.started = False
        self.elapsed_s = 0.0
        self.count = 0

    def elapsed(self, reset=True):
        val = self.elapsed_s
        if self.started:
            val += time.perf_counter() - self._start
        if reset:
            self.reset()
        return val

    def mean(self):
        return self.elapsed_s / max(self.count, 1)


class SynchronizedWallClockTimer:
    def __init__(self):
        self.timers = {}

    def __call__(self, name) -> _Timer:
        if name not in self.timers:
            self.timers[name] = _Timer(name)
        return self.timers[name]

    def log(self, names, normalizer=1.0, reset=True, ranks=None):
        parts = []
        for name in names:
            if name in self.timers:
                val = self.timers[name].elapsed(reset=reset) * 1000.0 / normalizer
                parts.append(f"{name}: {val:.2f}ms")
        if parts:
            log_dist(" | ".join(parts), ranks=ranks)


class ThroughputTimer:
    def __init__(self, batch_size, start_step=2, steps_per_output=50):
        self.batch_size = max(batch_size, 1)
        self.start_step = start_step
        self.steps_per_output = steps_per_output
        self.epoch_count = 0
        self.global_step_count = 0
        self.total_elapsed_time = 0.0
        self._start = None

    def start(self):
        self._start = time.perf_counter()

    def stop(self, global_step=True, report_speed=True):
        if self._start is None:
            return
        self.global_step_count += int(global_step)
        if self.global_step_count > self.start_step:
            self.total_elapsed_time += time.perf_counter() - self._start
            if report_speed and self.steps_per_output and \
                    self.global_step_count % self.steps_per_output == 0:
                log_dist(
                    f"step={self.global_step_count} "
                    f"samples/sec={self.avg_samples_per_sec():.2f}")
        self._start = None

    def avg_samples_per_sec(self):
        steps = self.global_step_count - self.start_step
        if steps <= 0 or self.total_elapsed_time == 0:
            return 0.0
        return steps * self.batch_size / self.total_elapsed_time
