"""Progressive Layer Dropping (reference:
deepspeed/runtime/progressive_layer_drop.py :40): theta(t) schedule that
layers consult to decide their keep probability during pretraining."""

import numpy as np


class ProgressiveLayerDrop:
    def __init__(self, theta: float = 0.5, gamma: float = 0.001):
        self.theta = theta
        self.gamma = gamma
        self.current_theta = 1.0

    def get_state(self):
        return {"progressive_layer_drop": True, "pld_theta": self.get_theta()}

    def get_theta(self) -> float:
        return self.current_theta

    def update_state(self, global_step: int):
        def _prob(x, g, p):
            return (1.0 - p) * np.exp(-g * x) + p
        self.current_theta = float(_prob(global_step, self.gamma, self.theta))
