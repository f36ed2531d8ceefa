"""zero.Init / GatheredParameters public API (reference:
deepspeed/runtime/zero/partition_parameters.py Init :824,
GatheredParameters :2121).

Design delta from the reference, on purpose: the reference shards every
parameter at construction time by monkey-patching ``nn.Module.__init__`` of
every subclass, because 40 GB A100s cannot hold a full model replica even
transiently. One MI355X node has 288 GB of HBM3E per GPU *and* TBs of host
DRAM, so this framework partitions at engine construction (stage3 builds
module-unit shards from the materialized module) and ``Init`` only needs to
keep the transient host copy small: modules constructed inside the context
are cast to the target dtype (bf16 halves host RAM for a 70B init) the
moment they finish ``__init__``. Checkpoint-load flows that never want full
weights should construct on the meta device and use
``engine.load_checkpoint`` instead.

``GatheredParameters`` delegates to the live ZeRO-3 optimizer's gather
context once an engine exists; before/without one it is a no-op (params
are still full).
"""

import functools
from typing import Optional

import torch
import torch.nn as nn

_active_stage3 = None  # set by ZeroStage3Optimizer.__init__


def register_stage3(opt):
    global _active_stage3
    _active_stage3 = opt


class Init:
    def __init__(self, module=None, config=None, dtype=None, enabled=True,
                 **_ignored):
        self.dtype = dtype or torch.bfloat16
        self.enabled = enabled
        self._orig_init = None
        if module is not None:  # eager form: cast an existing module
            module.to(self.dtype)

    def __enter__(self):
        if not self.enabled:
            return self
        self._orig_init = nn.Module.__init__
        target_dtype = self.dtype

        @functools.wraps(self._orig_init)
        def wrapped(mod, *args, **kwargs):
            Init._depth = getattr(Init, "_depth", 0) + 1
            try:
                self._orig_init(mod, *args, **kwargs)
            finally:
                Init._depth -= 1
            # cast only once construction fully finished (outermost module
            # sees children already cast; direct params cast here)
            for p in mod._parameters.values():
                if p is not None and p.is_floating_point():
                    p.data = p.data.to(target_dtype)
        nn.Module.__init__ = wrapped
        return self

    def __exit__(self, *exc):
        if self._orig_init is not None:
            nn.Module.__init__ = self._orig_init
        return False


class GatheredParameters:
    def __init__(self, params, modifier_rank: Optional[int] = None,
                 fwd_module=None, enabled: bool = True):
        self.enabled = enabled and _active_stage3 is not None
        self._inner = None
        if self.enabled:
            self._inner = _active_stage3.gathered_params(
                params, modifier_rank=modifier_rank)

    def __enter__(self):
        if self._inner is not None:
            self._inner.__enter__()
        return self

    def __exit__(self, *exc):
        if self._inner is not None:
            return self._inner.__exit__(*exc)
        return False
