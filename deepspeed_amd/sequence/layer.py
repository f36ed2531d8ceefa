"""Ulysses sequence parallelism — all-to-all head/sequence exchange.

Capability parity with the reference's ``deepspeed/sequence/layer.py``
(DistributedAttention :311, single_all_to_all :221), re-designed for the
MI355X node: the 8 GPUs form a full xGMI point-to-point mesh (7 links per
GPU), so an all-to-all uses every link simultaneously — it is the *best*
collective shape for this fabric, which makes Ulysses the preferred
long-context strategy here.

Data flow for one attention call with SP degree P:

    q,k,v  [b, s/P, H,  d]   --a2a-->   [b, s, H/P, d]
    local attention over the FULL sequence on H/P heads
    out    [b, s, H/P, d]   --a2a-->   [b, s/P, H,  d]

Each exchange is ONE ``all_to_all_single`` on a contiguous buffer laid out
so rank chunks are slices of dim 0 — no per-peer tensor lists.
"""

import torch
import torch.distributed as torch_dist

from .. import comm as dist


def _a2a_scatter_heads(x: torch.Tensor, group) -> torch.Tensor:
    """[b, s_local, H, d] -> [b, s, H/P, d] (scatter heads, gather sequence)."""
    P = dist.get_world_size(group)
    if P == 1:
        return x
    b, s_local, H, d = x.shape
    assert H % P == 0, f"heads {H} not divisible by sp world {P}"
    h = H // P
    # chunk heads per destination rank into dim 0
    t = x.reshape(b, s_local, P, h, d).permute(2, 0, 1, 3, 4).contiguous()
    out = torch.empty_like(t)
    dist.all_to_all_single(out, t, group=group)
    # dim 0 now indexes source rank = sequence chunk
    return (out.permute(1, 0, 2, 3, 4)         # [b, P, s_local, h, d]
               .reshape(b, P * s_local, h, d))


def _a2a_gather_heads(x: torch.Tensor, group) -> torch.Tensor:
    """[b, s, H/P, d] -> [b, s/P, H, d] (scatter sequence, gather heads)."""
    P = dist.get_world_size(group)
    if P == 1:
        return x
    b, s, h, d = x.shape
    assert s % P == 0, f"sequence {s} not divisible by sp world {P}"
    s_local = s // P
    t = x.reshape(b, P, s_local, h, d).permute(1, 0, 2, 3, 4).contiguous()
    out = torch.empty_like(t)
    dist.all_to_all_single(out, t, group=group)
    return (out.permute(1, 2, 0, 3, 4)         # [b, s_local, P, h, d]
               .reshape(b, s_local, P * h, d))


class _SeqAllToAll(torch.autograd.Function):
    """Differentiable sequence<->head all-to-all (reference layer.py:281)."""

    @staticmethod
    def forward(ctx, group, x, scatter_heads: bool):
        ctx.group = group
        ctx.scatter_heads = scatter_heads
        with torch.no_grad():
            return (_a2a_scatter_heads(x, group) if scatter_heads
                    else _a2a_gather_heads(x, group))

    @staticmethod
    def backward(ctx, grad):
        inv = (_a2a_gather_heads if ctx.scatter_heads else _a2a_scatter_heads)
        return None, inv(grad.contiguous(), ctx.group), None


class DistributedAttention(torch.nn.Module):
    """Wrap any local attention module for Ulysses sequence parallelism
    (reference sequence/layer.py:311).

    ``local_attn`` must accept (q, k, v, *args, **kwargs) with layout
    [b, seq, heads, head_dim] and return the same layout. The wrapper feeds
    it the full sequence with ``heads / sp_world`` heads per rank.
    """

    def __init__(self, local_attn, sequence_process_group,
                 scatter_idx: int = 2, gather_idx: int = 1):
        super().__init__()
        assert (scatter_idx, gather_idx) == (2, 1), \
            "layout is fixed to [b, s, H, d] (scatter heads, gather seq)"
        self.local_attn = local_attn
        self.spg = sequence_process_group

    def forward(self, query, key, value, *args, **kwargs):
        q = _SeqAllToAll.apply(self.spg, query, True)
        k = _SeqAllToAll.apply(self.spg, key, True)
        v = _SeqAllToAll.apply(self.spg, value, True)
        out = self.local_attn(q, k, v, *args, **kwargs)
        return _SeqAllToAll.apply(self.spg, out, False)


class UlyssesSPDataLoaderAdapter:
    """Shard each batch's sequence dimension across the SP group.

    Every SP rank must see the SAME batch (they jointly compute one sample's
    attention); this adapter broadcasts the batch from the SP-group leader
    and returns this rank's sequence slice. Labels shift happens before the
    split so cross-entropy stays local (reference: ulysses tutorial flow).
    """

    def __init__(self, group, seq_dim: int = 1):
        self.group = group
        self.seq_dim = seq_dim

    def shard(self, tensor: torch.Tensor) -> torch.Tensor:
        P = dist.get_world_size(self.group)
        if P == 1:
            return tensor
        r = dist.get_rank(self.group)
        src = torch_dist.get_global_rank(self.group, 0)
        dist.broadcast(tensor, src=src, group=self.group)
        s = tensor.size(self.seq_dim)
        assert s % P == 0, f"seq {s} not divisible by sp world {P}"
        return tensor.narrow(self.seq_dim, r * (s // P), s // P).contiguous()
