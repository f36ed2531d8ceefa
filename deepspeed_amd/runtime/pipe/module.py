"""PipelineModule — partition a flat layer list over pipeline stages
(reference: deepspeed/runtime/pipe/module.py PipelineModule :86,
LayerSpec :35, TiedLayerSpec :66, partitioning :393).

A model is expressed as an ordered list of callables / nn.Modules /
LayerSpecs whose forward is ``x -> x`` (tensor or tuple flows unchanged in
structure between layers). Each stage builds ONLY its own slice; tied
layers (e.g. embedding reused as lm_head) are replicated on every stage
that names the same key and their gradients all-reduce over a tie group
after each batch.
"""

import re
from typing import Callable, List, Optional

import torch
import torch.nn as nn

from ... import comm as dist
from ...utils.logging import log_dist
from .topology import PipelineParallelGrid


class LayerSpec:
    """Deferred layer construction: the class + args are recorded, the module
    is built only on the owning stage (reference module.py:35)."""

    def __init__(self, typename, *args, **kwargs):
        self.typename = typename
        self.args = args
        self.kwargs = kwargs

    def build(self):
        return self.typename(*self.args, **self.kwargs)

    def param_count(self) -> int:
        """Build once on the meta device to count parameters for the
        'parameters' partition method without allocating real storage."""
        try:
            with torch.device("meta"):
                m = self.build()
            return sum(p.numel() for p in m.parameters())
        except Exception:
            m = self.build()
            return sum(p.numel() for p in m.parameters())


class TiedLayerSpec(LayerSpec):
    """A LayerSpec replicated on every stage that uses the same ``key``;
    the instances share weights logically via post-batch gradient all-reduce
    over the tie group + init-time broadcast (reference module.py:66)."""

    def __init__(self, key, typename, *args, forward_fn: Optional[Callable] = None,
                 tied_weight_attr="weight", **kwargs):
        super().__init__(typename, *args, **kwargs)
        self.key = key
        self.forward_fn = forward_fn
        self.tied_weight_attr = tied_weight_attr


def _count_params(layer) -> int:
    if isinstance(layer, LayerSpec):
        return layer.param_count()
    if isinstance(layer, nn.Module):
        return sum(p.numel() for p in layer.parameters())
    return 0


def partition_balanced(weights: List[int], num_parts: int) -> List[int]:
    """Split ``weights`` into ``num_parts`` contiguous chunks minimizing the
    heaviest chunk (binary search over the bottleneck, greedy packing)."""
    n = len(weights)
    assert n >= num_parts, f"{n} layers < {num_parts} stages"
    prefix = [0]
    for w in weights:
        prefix.append(prefix[-1] + w)

    def parts_needed(cap):
        parts, cur = 1, 0
        for w in weights:
            if w > cap:
                return num_parts + 1
            if cur + w > cap:
                parts += 1
                cur = w
            else:
                cur += w
        return parts

    lo, hi = max(weights, default=0), max(prefix[-1], 1)
    while lo < hi:
        mid = (lo + hi) // 2
        if parts_needed(mid) <= num_parts:
            hi = mid
        else:
            lo = mid + 1
    cap = lo
    bounds = [0]
    cur = 0
    for i, w in enumerate(weights):
        parts_left = num_parts - (len(bounds) - 1)   # parts still open (incl. current)
        layers_left = n - i                          # layers not yet placed (incl. i)
        # close the current part if adding w would exceed cap, or if every
        # remaining layer is needed to give later parts one layer each
        if cur > 0 and (cur + w > cap or layers_left < parts_left):
            bounds.append(i)
            cur = 0
        cur += w
    while len(bounds) < num_parts:          # degenerate: pad with last layers
        bounds.append(n - (num_parts - len(bounds)))
    bounds.append(n)
    assert len(bounds) == num_parts + 1 and all(
        bounds[i] < bounds[i + 1] for i in range(num_parts)), bounds
    return bounds


class PipelineModule(nn.Module):
    def __init__(self, layers, num_stages: int = None, grid=None,
                 loss_fn: Optional[Callable] = None,
                 partition_method: str = "parameters",
                 activation_checkpoint_interval: int = 0,
                 seed_layers: bool = False, base_seed: int = 1234,
                 tp_size: int = 1):
        super().__init__()
        if not dist.is_initialized():
            dist.init_distributed()
        self.specs = list(layers)
        if grid is None:
            assert num_stages is not None, "need num_stages or grid"
            grid = PipelineParallelGrid(num_stages, tp_size=tp_size)
        self.grid = grid
        self.num_stages = grid.pipe_parallel_size
        self.stage_id = grid.stage_id
        self.loss_fn = loss_fn
        self.activation_checkpoint_interval = activation_checkpoint_interval

        self.parts = self._partition(partition_method)
        self.part_start = self.parts[self.stage_id]
        self.part_end = self.parts[self.stage_id + 1]

        self.forward_funcs: List = []
        self.tied_modules = nn.ModuleDict()
        self.tied_weight_attrs = {}
        # Local layers registered under their GLOBAL spec index (reference
        # pipe/module.py names layer files by global index too): state-dict
        # keys are then unique ACROSS stages, which same-PP resume doesn't
        # need but universal checkpoint conversion and cross-PP-degree
        # reshape do — local indices would collide stage 0's layer 0 with
        # stage 1's.
        self._layers = nn.ModuleDict()
        for idx in range(self.part_start, self.part_end):
            spec = self.specs[idx]
            if isinstance(spec, TiedLayerSpec):
                if spec.key not in self.tied_modules:
                    mod = spec.build()
                    if seed_layers:
                        torch.manual_seed(base_seed + idx)
                    self.tied_modules[spec.key] = mod
                mod = self.tied_modules[spec.key]
                self.tied_weight_attrs[spec.key] = spec.tied_weight_attr
                if spec.forward_fn is None:
                    self.forward_funcs.append(mod)
                else:
                    self.forward_funcs.append(
                        lambda x, m=mod, f=spec.forward_fn: f(m, x))
            elif isinstance(spec, LayerSpec):
                if seed_layers:
                    torch.manual_seed(base_seed + idx)
                mod = spec.build()
                self._layers[str(idx)] = mod
                self.forward_funcs.append(mod)
            elif isinstance(spec, nn.Module):
                self._layers[str(idx)] = spec
                self.forward_funcs.append(spec)
            elif callable(spec):
                self.forward_funcs.append(spec)
            else:
                raise TypeError(f"unsupported layer spec {type(spec)}")

        self._tie_groups = self._build_tie_groups()
        self._sync_tied_weights()
        log_dist(f"pipeline stage {self.stage_id}/{self.num_stages}: layers "
                 f"[{self.part_start}, {self.part_end}) "
                 f"params={sum(p.numel() for p in self.parameters())}")

    # ------------------------------------------------------------- partition
    def _partition(self, method: str) -> List[int]:
        n = len(self.specs)
        S = self.num_stages
        method = method.lower()
        if method == "uniform":
            weights = [1] * n
        elif method == "parameters":
            weights = [max(_count_params(l), 1) for l in self.specs]
        elif method.startswith("type:"):
            pat = method.split(":", 1)[1]
            weights = [1 if re.search(pat, type(l).__name__ if
                                      not isinstance(l, LayerSpec)
                                      else l.typename.__name__) else 0
                       for l in self.specs]
            if sum(weights) == 0:
                raise ValueError(f"no layers match type:{pat}")
        else:
            raise ValueError(f"unknown partition method {method}")
        return partition_balanced(weights, S)

    # ------------------------------------------------------------ tied layers
    def _tied_keys_per_stage(self):
        """key -> sorted list of stage ids that instantiate it."""
        keys = {}
        for stage in range(self.num_stages):
            lo, hi = self.parts[stage], self.parts[stage + 1]
            for spec in self.specs[lo:hi]:
                if isinstance(spec, TiedLayerSpec):
                    keys.setdefault(spec.key, set()).add(stage)
        return {k: sorted(v) for k, v in keys.items()}

    def _build_tie_groups(self):
        """One process group per tied key per pipe replica (all dp ids create
        the groups collectively; each rank keeps the ones it belongs to)."""
        tie_groups = {}
        tp_size = self.grid.tensor_parallel_size
        for key, stages in self._tied_keys_per_stage().items():
            if len(stages) < 2:
                continue
            for dp in range(self.grid.data_parallel_size):
                for tp in range(tp_size):
                    ranks = [(s * self.grid.data_parallel_size + dp) *
                             tp_size + tp for s in stages]
                    g = dist.new_group(ranks)
                    if self.grid.global_rank in ranks:
                        tie_groups[key] = (ranks, g)
        return tie_groups

    @torch.no_grad()
    def _sync_tied_weights(self):
        for key, (ranks, g) in self._tie_groups.items():
            mod = self.tied_modules[key]
            for p in mod.parameters():
                dist.broadcast(p.data, src=ranks[0], group=g)

    def allreduce_tied_weight_gradients(self):
        """All-reduce tied-weight grads over each tie group
        (reference pipe/engine.py:_exec_reduce_tied_grads:275)."""
        for key, (ranks, g) in self._tie_groups.items():
            mod = self.tied_modules[key]
            for p in mod.parameters():
                if p.grad is not None:
                    dist.all_reduce(p.grad, group=g)

    # ---------------------------------------------------------------- forward
    def forward(self, inputs):
        x = inputs
        ckpt_every = self.activation_checkpoint_interval
        if ckpt_every > 0 and self.training and torch.is_grad_enabled():
            from ..activation_checkpointing import checkpoint

            def run_span(start, end):
                def fn(x_):
                    for f in self.forward_funcs[start:end]:
                        x_ = f(x_) if torch.is_tensor(x_) or not \
                            isinstance(x_, tuple) else f(*x_)
                    return x_
                return fn
            i = 0
            while i < len(self.forward_funcs):
                j = min(i + ckpt_every, len(self.forward_funcs))
                if torch.is_tensor(x):
                    x = checkpoint(run_span(i, j), x)
                else:
                    x = checkpoint(run_span(i, j), *x)
                i = j
        else:
            for f in self.forward_funcs:
                x = f(x) if torch.is_tensor(x) or not isinstance(x, tuple) \
                    else f(*x)
        return x

    # --------------------------------------------------------------- helpers
    def is_first_stage(self):
        return self.stage_id == 0

    def is_last_stage(self):
        return self.stage_id == self.num_stages - 1
