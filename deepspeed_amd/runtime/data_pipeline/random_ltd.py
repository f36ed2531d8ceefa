"""Random-LTD — random layerwise token dropping (reference:
deepspeed/runtime/data_pipeline/data_routing/basic_layer.py
RandomLayerTokenDrop + scheduler; csrc/random_ltd token_sort/gather kernels
are replaced by torch.gather/scatter, which lower to single HIP gather
kernels on ROCm).

Middle layers process only a random subset of tokens; the subset grows on
a schedule until full length, cutting pretraining FLOPs ~2x at equal
quality (reference random-LTD paper).
"""

import torch
import torch.nn as nn


class RandomLTDScheduler:
    def __init__(self, total_layers: int, random_ltd_layer_num: int,
                 start_seq: int, max_seq: int, step_size: int,
                 schedule_steps: int):
        self.total_layers = total_layers
        self.random_ltd_layer_num = random_ltd_layer_num
        self.start_seq = start_seq
        self.max_seq = max_seq
        self.step_size = step_size
        self.schedule_steps = schedule_steps
        self.current_seq = start_seq
        self.global_step = 0

    def update_seq(self, global_step: int) -> int:
        self.global_step = global_step
        span = self.max_seq - self.start_seq
        frac = min(1.0, global_step / max(self.schedule_steps, 1))
        seq = self.start_seq + int(frac * span / self.step_size) * self.step_size
        self.current_seq = min(self.max_seq, seq)
        return self.current_seq

    def state_dict(self):
        return {"current_seq": self.current_seq,
                "global_step": self.global_step}

    def load_state_dict(self, sd):
        self.current_seq = sd["current_seq"]
        self.global_step = sd["global_step"]


class RandomLayerTokenDrop(nn.Module):
    """Wrap a decoder layer: sample ``scheduler.current_seq`` token
    positions (sorted, so causal order is preserved), run the layer on the
    subset, scatter results back into the passthrough sequence."""

    def __init__(self, layer: nn.Module, scheduler: RandomLTDScheduler):
        super().__init__()
        self.random_ltd_layer = layer
        self.scheduler = scheduler

    def forward(self, x: torch.Tensor, *args, **kwargs):
        B, S, Hd = x.shape
        keep = self.scheduler.current_seq
        if not self.training or keep >= S:
            return self.random_ltd_layer(x, *args, **kwargs)
        from ...ops.token_ops import token_gather, token_scatter
        idx = torch.stack([
            torch.randperm(S, device=x.device)[:keep].sort().values
            for _ in range(B)])                     # [B, keep] sorted
        sub = token_gather(x, idx)
        out = self.random_ltd_layer(sub, *args, **kwargs)
        if isinstance(out, tuple):
            out = out[0]
        return token_scatter(x, out, idx)


def convert_to_random_ltd(model: nn.Module, layers_attr: str,
                          scheduler: RandomLTDScheduler,
                          skip_first: int = 1, skip_last: int = 1):
    """Wrap the middle layers of ``model.<layers_attr>`` (a ModuleList)
    with RandomLayerTokenDrop (reference convert_to_random_ltd)."""
    obj = model
    for part in layers_attr.split("."):
        obj = getattr(obj, part)
    assert isinstance(obj, nn.ModuleList)
    n = len(obj)
    wrapped = 0
    for i in range(skip_first, n - skip_last):
        obj[i] = RandomLayerTokenDrop(obj[i], scheduler)
        wrapped += 1
    return wrapped
