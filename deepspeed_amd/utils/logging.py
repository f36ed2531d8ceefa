"""Rank-aware logging.

Capability parity with the reference's ``deepspeed/utils/logging.py``
(logger + log_dist), re-implemented for the MI355X framework.
"""

import logging
import os
import sys
from typing import Optional

_FORMAT = "[%(asctime)s] [%(levelname)s] [%(name)s:%(lineno)d] %(message)s"


def _create_logger(name: str = "deepspeed_amd", level=logging.INFO) -> logging.Logger:
    lg = logging.getLogger(name)
    if lg.handlers:
        return lg
    lg.setLevel(level)
    lg.propagate = False
    handler = logging.StreamHandler(stream=sys.stdout)
    handler.setFormatter(logging.Formatter(_FORMAT, datefmt="%Y-%m-%d %H:%M:%S"))
    lg.addHandler(handler)
    return lg


logger = _create_logger()


def _get_rank() -> int:
    try:
        import torch.distributed as dist
        if dist.is_available() and dist.is_initialized():
            return dist.get_rank()
    except Exception:
        pass
    return int(os.environ.get("RANK", 0))


def log_dist(message: str, ranks: Optional[list] = None, level=logging.INFO) -> None:
    """Log ``message`` only on the given ranks (default: rank 0)."""
    my_rank = _get_rank()
    ranks = ranks if ranks is not None else [0]
    if my_rank in ranks or -1 in ranks:
        logger.log(level, f"[Rank {my_rank}] {message}")


def warning_once(message: str, _seen=set()) -> None:
    if message not in _seen:
        _seen.add(message)
        logger.warning(message)
