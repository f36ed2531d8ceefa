"""rocTX range instrumentation (reference deepspeed/utils/nvtx.py:11).

On ROCm, ``torch.cuda.nvtx`` maps to rocTX markers, which rocprofv3
(--marker-trace) and Perfetto traces pick up — so profiles segment by
framework phase (forward / backward / reduce / gather / step) instead of
one anonymous kernel soup. Enabled when DS_AMD_ROCTX=1 (markers cost a
little on every call, so they are opt-in like the reference's
``enable_nvtx`` knob); ``instrument_w_nvtx`` is a no-op otherwise.
"""

import functools
import os

import torch

_ENABLED = os.environ.get("DS_AMD_ROCTX") == "1"


def roctx_enabled() -> bool:
    return _ENABLED


def enable_roctx(on: bool = True) -> None:
    global _ENABLED
    _ENABLED = on


def range_push(name: str) -> None:
    if _ENABLED:
        torch.cuda.nvtx.range_push(name)


def range_pop() -> None:
    if _ENABLED:
        torch.cuda.nvtx.range_pop()


class roctx_range:
    """Context manager form: ``with roctx_range("fetch"): ...``"""

    def __init__(self, name: str):
        self.name = name

    def __enter__(self):
        range_push(self.name)
        return self

    def __exit__(self, *exc):
        range_pop()
        return False


def instrument_w_nvtx(fn):
    """Decorator: wraps fn in a rocTX range named after it (reference
    @instrument_w_nvtx)."""
    qual = getattr(fn, "__qualname__", getattr(fn, "__name__", "fn"))

    @functools.wraps(fn)
    def wrapped(*args, **kwargs):
        if not _ENABLED:
            return fn(*args, **kwargs)
        torch.cuda.nvtx.range_push(qual)
        try:
            return fn(*args, **kwargs)
        finally:
            torch.cuda.nvtx.range_pop()

    return wrapped
