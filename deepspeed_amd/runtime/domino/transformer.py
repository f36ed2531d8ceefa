"""Domino — tensor parallelism with the TP all-reduce hidden behind compute
(reference: deepspeed/runtime/domino/transformer.py DominoTransformerLayer).

The TP row-parallel all-reduce after attention (and after the MLP) is on
the critical path. Domino splits each micro-batch into ``n_chunks`` along
the batch axis and interleaves: while chunk i's partial attention output is
in flight on the async all-reduce, chunk i+1's attention computes — on the
MI355X node the collective runs over xGMI concurrently with MFMA compute.

This layer owns the async handles explicitly instead of relying on a
compiler; numerics are identical to the unoverlapped layer (the tests
assert exact equality).
"""

from typing import List, Optional

import torch
import torch.nn as nn

from ... import comm as dist


class _AsyncAllReduce(torch.autograd.Function):
    """All-reduce whose wait is deferred: forward returns (tensor, handle)
    via module-side bookkeeping; backward is identity (row-parallel g)."""

    @staticmethod
    def forward(ctx, x, group, registry: list):
        x = x.contiguous()
        if dist.get_world_size(group) > 1:
            h = dist.all_reduce(x, group=group, async_op=True)
        else:
            h = None
        registry.append(h)
        return x

    @staticmethod
    def backward(ctx, grad):
        return grad, None, None


class DominoTransformerLayer(nn.Module):
    """Pre-norm transformer layer over TP-sharded attention/MLP modules.

    ``attn`` and ``mlp`` must produce PARTIAL (row-parallel) outputs, i.e.
    their final linear is sharded along input features WITHOUT the
    all-reduce — this layer inserts the deferred all-reduce itself.
    """

    def __init__(self, norm1: nn.Module, attn: nn.Module, norm2: nn.Module,
                 mlp: nn.Module, tp_group=None, n_chunks: int = 2):
        super().__init__()
        self.norm1 = norm1
        self.attn = attn
        self.norm2 = norm2
        self.mlp = mlp
        self.tp_group = tp_group
        self.n_chunks = n_chunks

    def forward(self, x: torch.Tensor, *attn_args, **attn_kwargs):
        chunks = list(torch.chunk(x, self.n_chunks, dim=0))
        handles: List = []
        partial: List[Optional[torch.Tensor]] = []

        # phase 1: attention per chunk, all-reduce launched immediately,
        # waited only when the chunk is next needed
        for c in chunks:
            a = self.attn(self.norm1(c), *attn_args, **attn_kwargs)
            partial.append(_AsyncAllReduce.apply(a, self.tp_group, handles))

        out_chunks = []
        mlp_handles: List = []
        mlp_partial: List[Optional[torch.Tensor]] = []
        for i, c in enumerate(chunks):
            if handles[i] is not None:
                handles[i].wait()
            h = c + partial[i]
            m = self.mlp(self.norm2(h))
            mlp_partial.append((h, _AsyncAllReduce.apply(m, self.tp_group,
                                                         mlp_handles)))
        for i, (h, m) in enumerate(mlp_partial):
            if mlp_handles[i] is not None:
                mlp_handles[i].wait()
            out_chunks.append(h + m)
        return torch.cat(out_chunks, dim=0)
