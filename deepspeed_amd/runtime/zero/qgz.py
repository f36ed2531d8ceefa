"""qgZ — quantized-gradient reduce for ZeRO-3 (reference: ZeRO++,
deepspeed/runtime/zero/stage3.py zero_quantized_gradients +
csrc/quantization swizzled_quant kernels).

Replaces the bf16 ``reduce_scatter_tensor`` of a unit's flat gradient with
an all-to-all of int8 groupwise-quantized blocks plus a local dequant-sum,
halving the wire bytes. Two shapes:

* **one-level** (single node / no hpz groups): quantize the W rank-blocks,
  ``all_to_all_single`` over the DP group, dequantize the W received
  partials and sum in fp32.
* **two-level** (hpz groups configured, i.e. multi-node): intra-node
  all-to-all first — after it each rank holds its node's partial sums for
  the global ranks sharing its local index — then a second quantized
  all-to-all across nodes. Each gradient crosses the slow inter-node links
  exactly once, quantized, and the intra-node hop rides xGMI. The
  reference fuses the block permutation ("swizzle") into its quant kernel;
  here the permute is a strided copy folded into the pre-quant reshape,
  and the groupwise quant itself is ops/csrc/quantize.hip on GPU.

Quant scales never cross rank-block boundaries (group size divides the
shard size), so per-block dequant is exact w.r.t. the quantizer.
"""

import torch

from ... import comm as dist
from ...ops.quantizer import quantize, dequantize


def _fit_group(numel: int, group_size: int) -> int:
    g = group_size
    while g > 2 and numel % g:
        g //= 2
    return g


def _quant_a2a(blocks: torch.Tensor, n_blocks: int, group, group_size: int,
               bits: int = 8):
    """all-to-all `n_blocks` equal flat blocks, quantized. Returns the
    received blocks dequantized to fp32, shape (n_blocks, block)."""
    block = blocks.numel() // n_blocks
    gs = _fit_group(block, group_size)
    q, s = quantize(blocks, gs, bits=bits)
    q_out = torch.empty_like(q)
    s_out = torch.empty_like(s)
    dist.all_to_all_single(q_out, q, group=group)
    dist.all_to_all_single(s_out, s, group=group)
    out = torch.empty(n_blocks, block, dtype=torch.float32,
                      device=blocks.device)
    qb = q.numel() // n_blocks          # quantized bytes per block
    sb = s.numel() // n_blocks          # scale groups per block
    for i in range(n_blocks):
        out[i] = dequantize(q_out[i * qb:(i + 1) * qb],
                            s_out[i * sb:(i + 1) * sb],
                            block, gs, bits=bits, dtype=torch.float32)
    return out


@torch.no_grad()
def quantized_reduce(grad: torch.Tensor, shard_size: int, dp_group,
                     intra_group=None, inter_group=None,
                     group_size: int = 2048, bits: int = 8) -> torch.Tensor:
    """Reduce-scatter `grad` (W * shard_size flat, already pre-divided)
    into this rank's shard via quantized all-to-all. Returns the shard in
    grad's dtype."""
    world = dist.get_world_size(dp_group)
    if world == 1:
        return grad[:shard_size].clone()
    if intra_group is None or dist.get_world_size(intra_group) in (1, world):
        parts = _quant_a2a(grad, world, dp_group, group_size, bits)
        return parts.sum(dim=0).to(grad.dtype)
    S = dist.get_world_size(intra_group)      # ranks per node
    N = world // S                            # nodes
    # swizzle: [n0s0 n0s1 .. n1s0 ..] -> for local peer j: its N node-blocks
    swz = grad.view(N, S, shard_size).transpose(0, 1).contiguous()
    parts = _quant_a2a(swz, S, intra_group, group_size, bits)
    partial = parts.view(S, N, shard_size).sum(dim=0)   # node-local sums
    parts2 = _quant_a2a(partial.reshape(-1).to(grad.dtype), N, inter_group,
                        group_size, bits)
    return parts2.sum(dim=0).to(grad.dtype)
