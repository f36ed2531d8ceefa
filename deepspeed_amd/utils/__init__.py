"""Reference-parity utils namespace (deepspeed/utils/__init__.py exports
logger, log_dist, groups, RepeatingLoader, see_memory_usage)."""

from .logging import logger, log_dist  # noqa: F401
from ..parallel import groups  # noqa: F401
from ..runtime.dataloader import RepeatingLoader  # noqa: F401
from ..runtime.utils import see_memory_usage  # noqa: F401
