"""TiledLinear (reference: deepspeed/runtime/zero/tiling.py, 296 LoC):
split one huge Linear into a grid of tile Linears so ZeRO-3 can partition,
gather and release each tile independently — bounds the transient
full-weight memory of very large projections (e.g. vocab heads) to one
tile instead of the whole matrix."""

import torch
import torch.nn as nn


def _split_sizes(total: int, parts: int):
    base = total // parts
    sizes = [base] * parts
    for i in range(total - base * parts):
        sizes[i] += 1
    return sizes


class TiledLinear(nn.Module):
    def __init__(self, in_features: int, out_features: int, bias: bool = True,
                 in_splits: int = 1, out_splits: int = 1,
                 linear_cls=nn.Linear, **linear_kwargs):
        super().__init__()
        assert 1 <= in_splits <= in_features
        assert 1 <= out_splits <= out_features
        self.in_features = in_features
        self.out_features = out_features
        self.in_splits = in_splits
        self.out_splits = out_splits
        self.in_sizes = _split_sizes(in_features, in_splits)
        self.out_sizes = _split_sizes(out_features, out_splits)
        self.linears = nn.ModuleList()
        for oi, osz in enumerate(self.out_sizes):
            for ii, isz in enumerate(self.in_sizes):
                # bias only on the first input tile of each output row
                self.linears.append(linear_cls(
                    isz, osz, bias=bias and ii == 0, **linear_kwargs))

    def _tile(self, oi, ii):
        return self.linears[oi * self.in_splits + ii]

    def forward(self, x):
        xs = torch.split(x, self.in_sizes, dim=-1)
        outs = []
        for oi in range(self.out_splits):
            acc = None
            for ii in range(self.in_splits):
                y = self._tile(oi, ii)(xs[ii])
                acc = y if acc is None else acc + y
            outs.append(acc)
        return torch.cat(outs, dim=-1)

    @torch.no_grad()
    def copy_params_from(self, other: nn.Linear):
        """Load from an untiled Linear (reference tiling.py
        copy_params_from)."""
        assert other.in_features == self.in_features
        assert other.out_features == self.out_features
        o0 = 0
        for oi, osz in enumerate(self.out_sizes):
            i0 = 0
            for ii, isz in enumerate(self.in_sizes):
                t = self._tile(oi, ii)
                t.weight.copy_(other.weight[o0:o0 + osz, i0:i0 + isz])
                if t.bias is not None and other.bias is not None:
                    t.bias.copy_(other.bias[o0:o0 + osz])
                i0 += isz
            o0 += osz
