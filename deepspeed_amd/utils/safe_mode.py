"""Safe-mode cross-rank consistency asserts (reference
partition_parameters.py:1241 assert_ints_same_as_other_ranks,
stage3.py:1312).

Distributed hangs are usually divergent control flow: one rank fetches a
different module order, sizes a bucket differently, or skips a collective.
With DS_AMD_SAFE_MODE=1 the framework cross-checks such integer sequences
across the group before the collectives that depend on them, turning a
hang into an immediate, attributed assertion. Off by default (each check
is an extra all-gather)."""

import os
from typing import Iterable, List

import torch

from .. import comm as dist

_ENABLED = os.environ.get("DS_AMD_SAFE_MODE") == "1"


def safe_mode_enabled() -> bool:
    return _ENABLED


def enable_safe_mode(on: bool = True) -> None:
    global _ENABLED
    _ENABLED = on


def assert_ints_same_as_other_ranks(ints: Iterable[int], group=None,
                                    what: str = "sequence") -> None:
    """All ranks in `group` must pass an identical list of ints. No-op
    unless safe mode is on (or called explicitly). Raises on rank(s) that
    diverge, naming the first differing position."""
    vals: List[int] = list(ints)
    if not dist.is_initialized() or dist.get_world_size(group) == 1:
        return
    t = torch.tensor(vals, dtype=torch.long)
    n = torch.tensor([t.numel()], dtype=torch.long)
    world = dist.get_world_size(group)
    lens = [torch.zeros_like(n) for _ in range(world)]
    dist.all_gather(lens, n, group=group)
    lens = [int(x.item()) for x in lens]
    if len(set(lens)) != 1:
        raise RuntimeError(
            f"safe-mode: ranks disagree on length of {what}: {lens}")
    gathered = [torch.zeros_like(t) for _ in range(world)]
    dist.all_gather(gathered, t, group=group)
    me = dist.get_rank(group)
    for r, other in enumerate(gathered):
        if not torch.equal(other, t):
            diff = (other != t).nonzero()[0].item()
            raise RuntimeError(
                f"safe-mode: {what} diverges between rank {me} and rank "
                f"{r} at position {diff}: {t[diff].item()} vs "
                f"{other[diff].item()}")


def checked(ints: Iterable[int], group=None, what: str = "sequence") -> None:
    """Run the cross-rank assert only when safe mode is enabled."""
    if _ENABLED:
        assert_ints_same_as_other_ranks(ints, group=group, what=what)
