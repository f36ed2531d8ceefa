"""Ulysses sequence parallelism (a2a head/sequence exchange over xGMI)."""

from .layer import (DistributedAttention, UlyssesSPDataLoaderAdapter,
                    _SeqAllToAll)

__all__ = ["DistributedAttention", "UlyssesSPDataLoaderAdapter"]
