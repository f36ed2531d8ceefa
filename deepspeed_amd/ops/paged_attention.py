"""Paged flash-decode attention wrapper (reference:
inference/v2/kernels/ragged_ops/blocked_flash). Consumes the PagedKVCache
pool + block tables directly — single-token decode never materializes the
padded [n, Hkv, L, D] views."""

import math
from typing import Optional

import torch

from ._loader import get_ext


def paged_decode_available(q: torch.Tensor, kpool: torch.Tensor) -> bool:
    G = q.size(1) // kpool.size(1)
    return (get_ext() is not None and q.is_cuda
            and q.dtype == torch.bfloat16 and q.size(-1) == 128
            and G in (1, 2, 4, 8))


def paged_decode_attention(q: torch.Tensor, kpool: torch.Tensor,
                           vpool: torch.Tensor, block_table: torch.Tensor,
                           lens: torch.Tensor, splits: Optional[int] = None,
                           scale: Optional[float] = None) -> torch.Tensor:
    """q [n, H, 128] bf16; kpool/vpool [nb, Hkv, BS, 128] bf16;
    block_table [n, max_blocks] int32; lens [n] int32 -> o [n, H, 128]."""
    ext = get_ext()
    scale = scale if scale is not None else 1.0 / math.sqrt(q.size(-1))
    if splits is None:
        # fill the chip: ~2048 workgroups wanted; grid = n * Hkv * splits
        nw = max(1, q.size(0) * kpool.size(1))
        splits = max(1, min(32, 2048 // nw))
    return ext.paged_decode(q.contiguous(), kpool, vpool,
                            block_table.contiguous().int(),
                            lens.contiguous().int(), splits, scale)
