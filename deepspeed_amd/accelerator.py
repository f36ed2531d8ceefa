"""Reference-path shim: ``from deepspeed_amd.accelerator import
get_accelerator`` (reference deepspeed/accelerator/real_accelerator.py).
The accelerator is the single ROCm/MI355X implementation in accel.py."""

from .accel import *  # noqa: F401,F403
from .accel import get_accelerator  # noqa: F401
