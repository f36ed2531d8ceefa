"""AutoTP — automatic tensor-parallel sharding of Linear layers for
inference (reference: deepspeed/module_inject/auto_tp.py AutoTP :192,
layers.py LinearLayer :370 / LinearAllreduce :300).

Column-parallel layers shard out_features (no comm); row-parallel layers
shard in_features and all-reduce the partial outputs over the TP group —
on the MI355X node that all-reduce runs over xGMI.

The generic policy shards by module-name suffix. For attention projections
the column split must respect head boundaries: q/k/v are sharded by whole
heads (their out_features are head multiples, so an even world split
preserves heads as long as heads % tp == 0, asserted by the engine).
"""

import re
from typing import Iterable, Optional

import torch
import torch.nn as nn

from .. import comm as dist


class LinearLayer(nn.Module):
    """Column-parallel linear: holds rows [r*out/P, (r+1)*out/P) of the
    weight; outputs this rank's slice of the features."""

    def __init__(self, weight: torch.Tensor, bias: Optional[torch.Tensor]):
        super().__init__()
        self.weight = nn.Parameter(weight, requires_grad=False)
        self.bias = nn.Parameter(bias, requires_grad=False) \
            if bias is not None else None

    def forward(self, x):
        return torch.nn.functional.linear(x, self.weight, self.bias)


class LinearAllreduce(nn.Module):
    """Row-parallel linear: holds columns of the weight; all-reduces the
    partial product over the TP group (reference layers.py:300)."""

    def __init__(self, weight: torch.Tensor, bias: Optional[torch.Tensor],
                 group):
        super().__init__()
        self.weight = nn.Parameter(weight, requires_grad=False)
        self.bias = nn.Parameter(bias, requires_grad=False) \
            if bias is not None else None
        self.group = group

    def forward(self, x):
        y = torch.nn.functional.linear(x, self.weight)
        if dist.get_world_size(self.group) > 1:
            dist.all_reduce(y, group=self.group)
        if self.bias is not None:
            y = y + self.bias
        return y


# module-name suffixes -> parallel style for common decoder architectures
COLUMN_PATTERNS = (r"q_proj$", r"k_proj$", r"v_proj$", r"gate_proj$",
                   r"up_proj$", r"w1$", r"w3$", r"c_attn$", r"wqkv$",
                   r"fc1$", r"c_fc$", r"gate_up_proj$")
ROW_PATTERNS = (r"o_proj$", r"down_proj$", r"w2$", r"attn\.c_proj$",
                r"mlp\.c_proj$", r"c_proj$", r"fc2$",
                r"dense$", r"out_proj$")
# fused projections whose out dim concatenates q|k|v: each rank must take
# its head slice from EACH third (reference auto_tp.py qkv handling)
FUSED_QKV_PATTERNS = (r"c_attn$", r"query_key_value$", r"wqkv$")


def _match(name: str, patterns: Iterable[str]) -> bool:
    return any(re.search(p, name) for p in patterns)


def _is_hf_conv1d(mod) -> bool:
    """transformers.pytorch_utils.Conv1D: weight [in, out], y = x@W + b
    (GPT-2 family). Detected structurally so transformers stays optional."""
    return (type(mod).__name__ == "Conv1D" and hasattr(mod, "weight")
            and hasattr(mod, "nf"))


def _fused_qkv_rows(out: int, tp_rank: int, tp_size: int) -> torch.Tensor:
    """Row indices of this rank's slice of a fused [3h] qkv out dim."""
    h = out // 3
    per = h // tp_size
    idx = []
    for third in range(3):
        base = third * h + tp_rank * per
        idx.append(torch.arange(base, base + per))
    return torch.cat(idx)


@torch.no_grad()
def _is_falcon_block(m: nn.Module) -> bool:
    return all(hasattr(m, a) for a in ("qkv", "dense", "mlp_fc",
                                       "mlp_proj", "num_heads",
                                       "num_kv_heads", "head_dim"))


def _shard_falcon_block(blk: nn.Module, tp_group, tp_rank: int,
                        tp_size: int) -> int:
    """Falcon's fused qkv concatenates [H*d | Hkv*d | Hkv*d] — NOT three
    equal thirds, so the generic fused-qkv split cannot apply. TP plan:
    split the q heads across ranks, REPLICATE the (grouped/multi-query)
    kv heads, row-shard `dense` over each rank's q-head columns, and
    column/row-shard the MLP pair. The block's cfg is replaced by a
    per-block copy with the local head count."""
    H, Hkv, d = blk.num_heads, blk.num_kv_heads, blk.head_dim
    assert H % tp_size == 0, f"falcon heads {H} not divisible by {tp_size}"
    hl = H // tp_size
    q_rows = torch.arange(tp_rank * hl * d, (tp_rank + 1) * hl * d)
    if Hkv % tp_size == 0:
        # GQA: each rank takes the kv heads that serve its q-head block
        # (splitting kv keeps the q->kv group mapping intact)
        kvl = Hkv // tp_size
        k0 = H * d + tp_rank * kvl * d
        v0 = (H + Hkv) * d + tp_rank * kvl * d
        kv_rows = torch.cat([torch.arange(k0, k0 + kvl * d),
                             torch.arange(v0, v0 + kvl * d)])
        new_kv = kvl
    else:
        # MQA (or kv not divisible): replicate every kv head; the local
        # q block maps onto the full (replicated) kv set
        assert Hkv == 1 or hl % Hkv == 0, \
            f"local heads {hl} not grouped by kv {Hkv}"
        kv_rows = torch.arange(H * d, (H + 2 * Hkv) * d)
        new_kv = Hkv
    rows = torch.cat([q_rows, kv_rows])
    W = blk.qkv.weight.data
    b = blk.qkv.bias.data if blk.qkv.bias is not None else None
    blk.qkv = LinearLayer(W[rows].clone(),
                          b[rows].clone() if b is not None else None)
    Wd = blk.dense.weight.data                     # [h, H*d]
    bd = blk.dense.bias.data if blk.dense.bias is not None else None
    blk.dense = LinearAllreduce(Wd[:, q_rows].clone(), bd, tp_group)
    Wf = blk.mlp_fc.weight.data                    # [4h, h]
    out = Wf.size(0)
    sl = slice(tp_rank * out // tp_size, (tp_rank + 1) * out // tp_size)
    bf = blk.mlp_fc.bias.data[sl].clone() \
        if blk.mlp_fc.bias is not None else None
    blk.mlp_fc = LinearLayer(Wf[sl].clone(), bf)
    Wp = blk.mlp_proj.weight.data                  # [h, 4h]
    bp = blk.mlp_proj.bias.data if blk.mlp_proj.bias is not None else None
    blk.mlp_proj = LinearAllreduce(Wp[:, sl].clone(), bp, tp_group)
    blk.num_heads = hl
    blk.num_kv_heads = new_kv
    return 4


def shard_model(model: nn.Module, tp_group, tp_rank: int, tp_size: int,
                column_patterns=COLUMN_PATTERNS, row_patterns=ROW_PATTERNS):
    """Replace matching nn.Linear modules with TP-sharded versions in place.

    Returns the number of modules sharded. Idempotent on already-replaced
    modules (they are no longer nn.Linear).
    """
    if tp_size == 1:
        return 0
    replaced = 0
    falcon_children = set()
    for m in model.modules():
        if _is_falcon_block(m):
            replaced += _shard_falcon_block(m, tp_group, tp_rank, tp_size)
            falcon_children.update(id(c) for c in m.modules() if c is not m)
    for parent_name, parent in list(model.named_modules()):
        for child_name, child in list(parent._modules.items()):
            if id(child) in falcon_children:
                continue
            is_conv1d = _is_hf_conv1d(child)
            if not isinstance(child, nn.Linear) and not is_conv1d:
                continue
            full = f"{parent_name}.{child_name}" if parent_name else child_name
            # Conv1D stores weight transposed ([in, out]); normalize to
            # linear orientation, shard, and keep serving in linear form
            W = child.weight.data.t().contiguous() if is_conv1d \
                else child.weight.data
            b = child.bias.data if child.bias is not None else None
            if _match(full, column_patterns):
                out = W.size(0)
                assert out % tp_size == 0, \
                    f"{full}: out_features {out} not divisible by tp {tp_size}"
                if _match(full, FUSED_QKV_PATTERNS) and out % 3 == 0:
                    rows = _fused_qkv_rows(out, tp_rank, tp_size)
                    new = LinearLayer(W[rows].clone(),
                                      b[rows].clone() if b is not None
                                      else None)
                else:
                    sl = slice(tp_rank * out // tp_size,
                               (tp_rank + 1) * out // tp_size)
                    new = LinearLayer(W[sl].clone(),
                                      b[sl].clone() if b is not None else None)
            elif _match(full, row_patterns):
                inp = W.size(1)
                assert inp % tp_size == 0, \
                    f"{full}: in_features {inp} not divisible by tp {tp_size}"
                sl = slice(tp_rank * inp // tp_size,
                           (tp_rank + 1) * inp // tp_size)
                # bias applied once, after the all-reduce
                new = LinearAllreduce(W[:, sl].clone(),
                                      b.clone() if b is not None else None,
                                      tp_group)
            else:
                continue
            parent._modules[child_name] = new
            replaced += 1
    return replaced


def shard_attention_heads(model: nn.Module, tp_rank: int, tp_size: int):
    """Patch head counts on attention modules that track them
    (reference auto_tp.py shard head counts via mp sizing)."""
    if tp_size == 1:
        return
    for mod in model.modules():
        if _is_falcon_block(mod):
            # already sharded by _shard_falcon_block: q heads split,
            # kv heads REPLICATED — the generic halving does not apply
            continue
        for attr in ("num_heads", "num_kv_heads", "num_attention_heads",
                     "num_key_value_heads"):
            n = getattr(mod, attr, None)
            if isinstance(n, int) and n > 0:
                assert n % tp_size == 0, \
                    f"{attr}={n} not divisible by tp={tp_size}"
                setattr(mod, attr, n // tp_size)
        # GPT-2-style fused attention: split_size/embed_dim size the
        # c_attn output split and the head reshape — scale them with tp
        if hasattr(mod, "c_attn"):
            for attr in ("split_size", "embed_dim"):
                n = getattr(mod, attr, None)
                if isinstance(n, int) and n > 0 and n % tp_size == 0:
                    setattr(mod, attr, n // tp_size)
