"""Domino TP-overlap layer tests (reference contract:
tests/unit/runtime/test_domino.py-equivalent): chunked/overlapped execution
must be numerically identical to the plain layer, fwd and bwd."""

import torch
import torch.nn as nn

from .common import run_distributed


class _PartialRowLinearMLP(nn.Module):
    """Row-parallel-style module producing PARTIAL outputs (no allreduce)."""

    def __init__(self, h, seed):
        super().__init__()
        torch.manual_seed(seed)
        self.up = nn.Linear(h, 2 * h, bias=False)
        self.down = nn.Linear(2 * h, h, bias=False)

    def forward(self, x):
        return self.down(torch.nn.functional.gelu(self.up(x)))


def _domino_worker(rank, world):
    from deepspeed_amd.runtime.domino import DominoTransformerLayer
    import torch.distributed as td
    g = td.group.WORLD
    h = 32
    torch.manual_seed(3)
    # rank-dependent weights emulate TP shards producing partial sums
    attn = _PartialRowLinearMLP(h, seed=100 + rank)
    mlp = _PartialRowLinearMLP(h, seed=200 + rank)
    n1, n2 = nn.LayerNorm(h), nn.LayerNorm(h)

    layer = DominoTransformerLayer(n1, attn, n2, mlp, tp_group=g, n_chunks=2)
    torch.manual_seed(7)
    x = torch.randn(4, 6, h, requires_grad=True)
    out = layer(x)
    out.sum().backward()
    grad_overlapped = x.grad.clone()

    # reference: same math, synchronous allreduce, no chunking
    def sync_ref(xr):
        a = attn(n1(xr))
        td.all_reduce(a)
        hmid = xr + a
        m = mlp(n2(hmid))
        td.all_reduce(m)
        return hmid + m

    x2 = x.detach().clone().requires_grad_(True)
    ref = sync_ref(x2)
    torch.testing.assert_close(out, ref, rtol=1e-6, atol=1e-6)
    ref.sum().backward()
    # backward grads equal too: the g-function is identity in backward and
    # each rank's dgrad flows through its own shard
    torch.testing.assert_close(grad_overlapped, x2.grad, rtol=1e-5, atol=1e-6)


def test_domino_matches_sync():
    run_distributed(_domino_worker, world_size=2)
