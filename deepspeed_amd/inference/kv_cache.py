"""KV cache for autoregressive decoding.

Capability parity with the reference's inference KV handling
(csrc/transformer/inference/includes/inference_context.h — persistent
per-layer workspace; inference/v2/ragged paged KV). This implementation is
a contiguous per-layer cache sized once for the generation
([B, kv_heads, max_seq, head_dim] per layer, bf16): on a 288 GB MI355X a
Llama-3-8B cache at 8k context is ~1 GB per 16 sequences, so paging buys
nothing until far larger batch x context than a single node serves — the
contiguous layout keeps SDPA reads fully coalesced instead.
"""

from typing import List, Tuple

import torch


class StaticKVCache:
    """Pre-allocated per-layer K/V with an append cursor."""

    def __init__(self, num_layers: int, batch: int, kv_heads: int,
                 max_seq: int, head_dim: int, dtype=torch.bfloat16,
                 device="cuda"):
        self.max_seq = max_seq
        self.cur_len = 0
        shape = (batch, kv_heads, max_seq, head_dim)
        self.k: List[torch.Tensor] = [
            torch.empty(shape, dtype=dtype, device=device)
            for _ in range(num_layers)]
        self.v: List[torch.Tensor] = [
            torch.empty(shape, dtype=dtype, device=device)
            for _ in range(num_layers)]
        self._pending = 0

    def update(self, layer_idx: int, k: torch.Tensor,
               v: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
        """Append [B, H_kv, S_new, D] at the cursor; returns views of the
        full prefix [B, H_kv, cur+S_new, D]. The cursor advances once per
        model step (advance() after the last layer)."""
        s_new = k.size(2)
        start = self.cur_len
        end = start + s_new
        assert end <= self.max_seq, f"KV cache overflow ({end}>{self.max_seq})"
        self.k[layer_idx][:, :, start:end].copy_(k)
        self.v[layer_idx][:, :, start:end].copy_(v)
        self._pending = s_new
        return (self.k[layer_idx][:, :, :end], self.v[layer_idx][:, :, :end])

    def advance(self):
        self.cur_len += self._pending
        self._pending = 0

    def reset(self):
        self.cur_len = 0
        self._pending = 0
