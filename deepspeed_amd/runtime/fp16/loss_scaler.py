"""Static and dynamic loss scaling for fp16 training.

Capability parity with the reference's ``deepspeed/runtime/fp16/loss_scaler.py``
(LossScaler :67, DynamicLossScaler :91).
"""

import torch

from ...utils.logging import logger


class LossScalerBase:
    def __init__(self, scale: float):
        self.cur_scale = float(scale)
        self.dynamic = False

    @property
    def loss_scale(self) -> float:
        return self.cur_scale

    def scale_gradient(self, module, grad_in, grad_out):
        return tuple(self.loss_scale * g for g in grad_in)

    def update_scale(self, overflow: bool):
        pass

    def backward(self, loss, retain_graph=False):
        (loss * self.loss_scale).backward(retain_graph=retain_graph)

    def state_dict(self):
        return {"cur_scale": self.cur_scale}

    def load_state_dict(self, sd):
        self.cur_scale = sd["cur_scale"]


class LossScaler(LossScalerBase):
    """Static loss scaling."""

    def __init__(self, scale=1.0):
        super().__init__(scale)


class DynamicLossScaler(LossScalerBase):
    """Doubles the scale every ``scale_window`` overflow-free steps; halves on
    overflow (with hysteresis)."""

    def __init__(self, init_scale=2 ** 16, scale_factor=2.0, scale_window=1000,
                 min_scale=1.0, delayed_shift=1, consecutive_hysteresis=False):
        super().__init__(init_scale)
        self.dynamic = True
        self.scale_factor = scale_factor
        self.scale_window = scale_window
        self.min_scale = min_scale
        self.delayed_shift = delayed_shift
        self.cur_hysteresis = delayed_shift
        self.consecutive_hysteresis = consecutive_hysteresis
        self.last_overflow_iter = -1
        self.cur_iter = 0

    def update_scale(self, overflow: bool):
        if overflow:
            if self.delayed_shift == 1 or self.cur_hysteresis == 1:
                self.cur_scale = max(self.cur_scale / self.scale_factor, self.min_scale)
                logger.info(f"overflow: reducing loss scale to {self.cur_scale}")
            else:
                self.cur_hysteresis -= 1
            self.last_overflow_iter = self.cur_iter
        else:
            if self.consecutive_hysteresis:
                self.cur_hysteresis = self.delayed_shift
            if (self.cur_iter - self.last_overflow_iter) % self.scale_window == 0 and \
                    self.cur_iter > self.last_overflow_iter:
                if not self.consecutive_hysteresis:
                    self.cur_hysteresis = self.delayed_shift
                self.cur_scale *= self.scale_factor
        self.cur_iter += 1

    def state_dict(self):
        return {"cur_scale": self.cur_scale, "cur_iter": self.cur_iter,
                "last_overflow_iter": self.last_overflow_iter,
                "cur_hysteresis": self.cur_hysteresis}

    def load_state_dict(self, sd):
        self.cur_scale = sd["cur_scale"]
        self.cur_iter = sd.get("cur_iter", 0)
        self.last_overflow_iter = sd.get("last_overflow_iter", -1)
        self.cur_hysteresis = sd.get("cur_hysteresis", self.delayed_shift)


def create_loss_scaler(fp16_config) -> LossScalerBase:
    """Build from FP16Config (loss_scale==0 => dynamic)."""
    if fp16_config.loss_scale and fp16_config.loss_scale > 0:
        return LossScaler(scale=fp16_config.loss_scale)
    return DynamicLossScaler(init_scale=2.0 ** fp16_config.initial_scale_power,
                             scale_window=fp16_config.loss_scale_window,
                             min_scale=fp16_config.min_loss_scale,
                             delayed_shift=fp16_config.hysteresis)


@torch.no_grad()
def has_inf_or_nan(tensor: torch.Tensor) -> bool:
    if tensor.numel() == 0:
        return False
    s = tensor.float().sum()
    return bool(torch.isinf(s) or torch.isnan(s))
