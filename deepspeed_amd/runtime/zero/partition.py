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
    """Context for constructing models under ZeRO-3.

    Two modes:
    * default: registered floating params are cast to ``dtype`` the moment
      they register (keeps the transient host copy small; partitioning
      happens at engine construction — the 288 GB HBM design stance).
    * ``remote_device="meta"``: params are registered on the META device —
      NO weight memory is allocated anywhere at construction, so models
      whose full weights exceed device+host memory can still be built.
      The stage-3 optimizer materializes each module unit transiently at
      partition time (``reset_parameters()`` per module, rank-0 broadcast)
      and keeps only the local shard. Reference analogue:
      partition_parameters.py Init :824 (construction-time sharding) +
      remote_device handling :1551.
    """

    def __init__(self, module=None, config=None, dtype=None, enabled=True,
                 remote_device=None, meta_device=False, **_ignored):
        self.dtype = dtype or torch.bfloat16
        self.enabled = enabled
        self.meta = meta_device or remote_device == "meta"
        self._orig_init = None
        if module is not None:  # eager form: cast an existing module
            if self.meta:
                module.to_empty(device="meta")
            else:
                module.to(self.dtype)

    def __enter__(self):
        if not self.enabled:
            return self
        # Parameters are REGISTERED (not constructed) through
        # Module.__setattr__ / register_parameter — always after the
        # subclass's super().__init__() has returned — so the cast must
        # hook registration, not nn.Module.__init__.
        self._orig_setattr = nn.Module.__setattr__
        self._orig_register = nn.Module.register_parameter
        target_dtype = self.dtype

        meta = self.meta

        @functools.wraps(self._orig_setattr)
        def wrapped_setattr(mod, name, value):
            if isinstance(value, nn.Parameter) and value.is_floating_point() \
                    and not value.is_meta:
                if meta:
                    value = nn.Parameter(
                        torch.empty(value.shape, dtype=target_dtype,
                                    device="meta"),
                        requires_grad=value.requires_grad)
                else:
                    value.data = value.data.to(target_dtype)
            self._orig_setattr(mod, name, value)

        @functools.wraps(self._orig_register)
        def wrapped_register(mod, name, param):
            if isinstance(param, nn.Parameter) and \
                    param.is_floating_point() and not param.is_meta:
                if meta:
                    param = nn.Parameter(
                        torch.empty(param.shape, dtype=target_dtype,
                                    device="meta"),
                        requires_grad=param.requires_grad)
                else:
                    param.data = param.data.to(target_dtype)
            self._orig_register(mod, name, param)

        nn.Module.__setattr__ = wrapped_setattr
        nn.Module.register_parameter = wrapped_register
        return self

    def __exit__(self, *exc):
        if getattr(self, "_orig_setattr", None) is not None:
            nn.Module.__setattr__ = self._orig_setattr
            nn.Module.register_parameter = self._orig_register
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
