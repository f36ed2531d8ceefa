"""Knowledge distillation + staged compression scheduling (reference
deepspeed/compression: layer_reduction / distillation config in
compress.py + the schedule_offset machinery in basic_layer.py).

Three pieces the round-1 miniature lacked:

* ``KDLoss`` — the standard distillation objective: KL(student/T ||
  teacher/T) * T^2 blended with the hard-label loss, plus optional
  intermediate-layer MSE terms (the reference's kd_loss + inter-layer
  distillation used by compressed-student training).
* ``LayerReduction`` — build a depth-reduced student from a teacher by
  keeping a subset of transformer layers (reference layer_reduction:
  teacher_layer list) with weight copy.
* ``CompressionScheduler`` — activates each compression method at its
  configured step offset (reference schedule_offset / schedule_offset_end
  on every *_quantization / *_pruning group) so QAT/pruning ramp in
  mid-training instead of from step 0.
"""

from typing import Dict, List, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F


class KDLoss(nn.Module):
    """loss = alpha * hard + (1-alpha) * T^2 * KL(student || teacher)
    (+ beta * mean MSE over paired hidden states when provided)."""

    def __init__(self, temperature: float = 2.0, alpha: float = 0.5,
                 beta: float = 0.0):
        super().__init__()
        self.T = temperature
        self.alpha = alpha
        self.beta = beta

    def forward(self, student_logits, teacher_logits, hard_loss=None,
                student_states: Optional[List[torch.Tensor]] = None,
                teacher_states: Optional[List[torch.Tensor]] = None):
        T = self.T
        soft = F.kl_div(
            F.log_softmax(student_logits.float() / T, dim=-1),
            F.softmax(teacher_logits.float().detach() / T, dim=-1),
            reduction="batchmean") * (T * T)
        loss = soft if hard_loss is None else \
            self.alpha * hard_loss + (1 - self.alpha) * soft
        if self.beta and student_states and teacher_states:
            inter = torch.stack(
                [F.mse_loss(s.float(), t.float().detach())
                 for s, t in zip(student_states, teacher_states)]).mean()
            loss = loss + self.beta * inter
        return loss


def build_reduced_student(teacher: nn.Module, keep_layers: List[int],
                          layers_attr: str = "model.layers") -> nn.Module:
    """Depth-reduce a copy of `teacher` by keeping `keep_layers` (teacher
    indices) — the reference's layer_reduction with teacher_layer mapping.
    Returns the student (deep copy; teacher untouched)."""
    import copy
    student = copy.deepcopy(teacher)
    obj = student
    parts = layers_attr.split(".")
    for p in parts[:-1]:
        obj = getattr(obj, p)
    layer_list = getattr(obj, parts[-1])
    assert all(0 <= i < len(layer_list) for i in keep_layers), \
        (keep_layers, len(layer_list))
    setattr(obj, parts[-1],
            nn.ModuleList([layer_list[i] for i in keep_layers]))
    return student


class CompressionScheduler:
    """Drives schedule offsets: each (module, method, kwargs, offset[,end])
    entry activates when `step(global_step)` crosses its offset, and
    freezes (fix_*) past its end. The engine calls `scheduler.step(n)`
    once per optimizer step."""

    def __init__(self):
        self.entries: List[Dict] = []
        self._done = set()
        self._frozen = set()

    def register(self, module, method: str, offset: int = 0,
                 end: Optional[int] = None, **kwargs):
        self.entries.append(dict(module=module, method=method,
                                 offset=offset, end=end, kwargs=kwargs))

    def step(self, global_step: int):
        for i, e in enumerate(self.entries):
            if i not in self._done and global_step >= e["offset"]:
                getattr(e["module"], f"enable_{e['method']}")(**e["kwargs"])
                self._done.add(i)
            if (i in self._done and i not in self._frozen
                    and e["end"] is not None and global_step >= e["end"]):
                fix = {"weight_quantization": "fix_weight_quantization",
                       "sparse_pruning": "fix_sparsity",
                       "row_pruning": "fix_sparsity"}.get(e["method"])
                if fix and hasattr(e["module"], fix):
                    getattr(e["module"], fix)()
                self._frozen.add(i)

    def state_dict(self):
        return {"done": sorted(self._done), "frozen": sorted(self._frozen)}

    def load_state_dict(self, sd):
        self._done = set(sd["done"])
        self._frozen = set(sd["frozen"])
