"""Block-sparse attention (reference: deepspeed/ops/sparse_attention —
Triton block-sparse matmul/softmax with Fixed/BigBird/BSLongformer/
Variable sparsity configs).

MI355X round-1 implementation: the block LAYOUT machinery (the API users
configure) with attention computed through SDPA using the layout expanded
to a dense mask — numerically identical to the reference's kernels, O(S^2)
compute. A gather-based HIP block kernel that realizes the FLOP savings is
the planned follow-up; the layouts and module interface are stable.
"""

import math
from typing import Optional

import torch
import torch.nn.functional as F


class SparsityConfig:
    def __init__(self, num_heads: int, block: int = 16):
        self.num_heads = num_heads
        self.block = block

    def make_layout(self, seq_len: int) -> torch.Tensor:
        raise NotImplementedError

    def _empty(self, seq_len):
        assert seq_len % self.block == 0, \
            f"seq {seq_len} not divisible by block {self.block}"
        n = seq_len // self.block
        return torch.zeros(self.num_heads, n, n, dtype=torch.bool), n


class DenseSparsityConfig(SparsityConfig):
    def make_layout(self, seq_len):
        l, n = self._empty(seq_len)
        return l | True


class FixedSparsityConfig(SparsityConfig):
    """Fixed pattern (reference sparsity_config.py Fixed): local window of
    ``num_local_blocks`` + every ``num_global_blocks``-th block attended
    globally; optionally causal."""

    def __init__(self, num_heads: int, block: int = 16,
                 num_local_blocks: int = 4, num_global_blocks: int = 1,
                 attention: str = "bidirectional"):
        super().__init__(num_heads, block)
        self.num_local_blocks = num_local_blocks
        self.num_global_blocks = num_global_blocks
        self.causal = attention == "unidirectional"

    def make_layout(self, seq_len):
        l, n = self._empty(seq_len)
        for i in range(n):
            w0 = (i // self.num_local_blocks) * self.num_local_blocks
            for j in range(w0, min(w0 + self.num_local_blocks, n)):
                l[:, i, j] = True
            for j in range(0, n, self.num_global_blocks):
                # global columns: last block of each local window
                g = min(j + self.num_local_blocks, n) - 1 \
                    if self.num_global_blocks == 1 else j
                l[:, i, g % n] = True
        if self.causal:
            tri = torch.tril(torch.ones(n, n, dtype=torch.bool))
            l &= tri
        return l


class BigBirdSparsityConfig(SparsityConfig):
    """random + sliding-window + global blocks (reference BigBird)."""

    def __init__(self, num_heads: int, block: int = 16,
                 num_random_blocks: int = 1, num_sliding_window_blocks: int = 3,
                 num_global_blocks: int = 1, seed: int = 0):
        super().__init__(num_heads, block)
        self.num_random_blocks = num_random_blocks
        self.num_sliding = num_sliding_window_blocks
        self.num_global = num_global_blocks
        self.seed = seed

    def make_layout(self, seq_len):
        l, n = self._empty(seq_len)
        half = self.num_sliding // 2
        g = torch.Generator().manual_seed(self.seed)
        for i in range(n):
            for j in range(max(0, i - half), min(n, i + half + 1)):
                l[:, i, j] = True
            l[:, i, :self.num_global] = True
            l[:, :self.num_global, i] = True
            for h in range(self.num_heads):
                for j in torch.randint(0, n, (self.num_random_blocks,),
                                       generator=g).tolist():
                    l[h, i, j] = True
        return l


class SparseSelfAttention(torch.nn.Module):
    """Applies attention under a block-sparse layout
    (reference sparse_self_attention.py). q,k,v: [B, H, S, D]."""

    def __init__(self, sparsity_config: SparsityConfig, max_seq_length=2048):
        super().__init__()
        self.config = sparsity_config
        self._layouts = {}

    def _mask(self, seq_len, device):
        if seq_len not in self._layouts:
            layout = self.config.make_layout(seq_len)
            mask = layout.repeat_interleave(self.config.block, dim=1) \
                         .repeat_interleave(self.config.block, dim=2)
            self._layouts[seq_len] = mask
        return self._layouts[seq_len].to(device)

    def forward(self, query, key, value, attn_mask: Optional[torch.Tensor] = None):
        B, H, S, D = query.shape
        mask = self._mask(S, query.device).unsqueeze(0)  # [1, H, S, S]
        if attn_mask is not None:
            mask = mask & attn_mask.bool()
        return F.scaled_dot_product_attention(query, key, value,
                                              attn_mask=mask)
