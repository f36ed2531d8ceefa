"""hipGraph-captured decode (reference inference/engine.py:494 CUDA-graph
path, re-designed for capture-safety on ROCm).

The eager decode step launches ~8 kernels per transformer layer; at
batch 1-8 the launch+python overhead dominates single-token latency.
``torch.cuda.CUDAGraph`` (hipGraph on ROCm) replays the whole step as one
graph launch — but capture requires static shapes AND static addresses,
which the eager ``StaticKVCache`` path breaks by slicing the cache to the
current length (shape changes every token).

``GraphKVCache`` keeps attention static-shape: ``update`` scatters the new
token's k/v at a position read from a DEVICE counter tensor and returns
the FULL cache buffers plus an additive mask derived from the counter —
dynamic values, static shapes, so the whole decode step captures.
``graph_generate`` then: prefills eagerly, warms the step up twice,
captures it once, and replays per token (host work per token = one H2D
token copy + argmax/sampling on the static logits buffer)."""

from typing import Optional

import torch


class GraphKVCache:
    """Capture-safe KV cache: fixed-size buffers, device-side length."""

    def __init__(self, n_layers, batch, kv_heads, max_seq, head_dim,
                 dtype=torch.bfloat16, device="cuda"):
        self.k = torch.zeros(n_layers, batch, kv_heads, max_seq, head_dim,
                             dtype=dtype, device=device)
        self.v = torch.zeros_like(self.k)
        self.max_seq = max_seq
        # device counter: tokens already in the cache
        self.len_t = torch.zeros((), dtype=torch.long, device=device)
        self.cur_len = 0           # host mirror (prefill/loop control only)
        self.last_mask = None
        self._arange = torch.arange(max_seq, device=device)

    @property
    def seq_len(self):
        return self.cur_len

    def update(self, layer_idx, k, v):
        """k, v: [B, H, s, D] -> full-length buffers + additive mask."""
        s = k.size(2)
        if s > 1:  # prefill (eager, not captured)
            self.k[layer_idx, :, :, :s] = k
            self.v[layer_idx, :, :, :s] = v
            # causal prefill mask over the static length
            q_pos = self._arange[:s].view(s, 1)
            kv_ok = self._arange.view(1, -1) <= q_pos
            self.last_mask = torch.where(
                kv_ok, 0.0, float("-inf")).to(k.dtype)
        else:  # single-token decode (capture-safe)
            idx = self.len_t.reshape(1)
            self.k[layer_idx].index_copy_(2, idx, k)
            self.v[layer_idx].index_copy_(2, idx, v)
            kv_ok = self._arange.view(1, -1) <= self.len_t
            self.last_mask = torch.where(
                kv_ok, 0.0, float("-inf")).to(k.dtype).view(1, 1, 1, -1)
        return self.k[layer_idx], self.v[layer_idx]

    def advance(self, n=1):
        # host mirror only; the device counter advances inside the
        # captured region (see graph_generate) so replays self-advance
        self.cur_len += n


@torch.no_grad()
def graph_generate(module, input_ids, *, n_layers, kv_heads, head_dim,
                   max_seq, dtype, max_new_tokens=32, do_sample=False,
                   temperature=1.0, top_k=0, top_p=1.0,
                   eos_token_id: Optional[int] = None, warmup=2):
    """hipGraph-replayed greedy/sampled decode. Same contract as
    kv_generate (engine.py) — prompt [B, S] -> [B, S + new]."""
    from .engine import _select_token

    device = input_ids.device
    B, S = input_ids.shape
    total = min(S + max_new_tokens, max_seq)
    kv = GraphKVCache(n_layers, B, kv_heads, total, head_dim, dtype=dtype,
                      device=device)

    # ---- eager prefill
    positions = torch.arange(S, device=device,
                             dtype=torch.int32).expand(B, S).contiguous()
    logits = module(input_ids, positions=positions, kv_cache=kv)
    kv.len_t.fill_(S)
    kv.advance(S)
    next_tok = _select_token(logits[:, -1], do_sample, temperature, top_k,
                             top_p)
    out = torch.cat([input_ids, next_tok], dim=1)

    # ---- static buffers for the captured step
    in_buf = next_tok.clone()

    def step():
        pos = kv.len_t.to(torch.int32).repeat(B).view(B, 1)
        lg = module(in_buf, positions=pos, kv_cache=kv)
        kv.len_t.add_(1)  # in-graph: each replay self-advances
        return lg[:, -1]

    # warmup on a side stream (cements allocator blocks), then capture
    side = torch.cuda.Stream()
    side.wait_stream(torch.cuda.current_stream())
    saved_len = kv.cur_len
    with torch.cuda.stream(side):
        for _ in range(warmup):
            kv.len_t.fill_(saved_len)
            static_logits = step()
    torch.cuda.current_stream().wait_stream(side)
    kv.len_t.fill_(saved_len)
    graph = torch.cuda.CUDAGraph()
    with torch.cuda.graph(graph):
        static_logits = step()
    # capture RECORDS the step without executing it — replay once so the
    # first decode token's k/v actually lands in the cache
    graph.replay()
    kv.advance()
    next_tok = _select_token(static_logits, do_sample, temperature, top_k,
                             top_p)
    out = torch.cat([out, next_tok], dim=1)
    finished = torch.zeros(B, dtype=torch.bool, device=device)

    while out.size(1) < total:
        if eos_token_id is not None:
            finished |= next_tok.squeeze(1) == eos_token_id
            if bool(finished.all()):
                break
        in_buf.copy_(next_tok)
        graph.replay()
        kv.advance()
        next_tok = _select_token(static_logits, do_sample, temperature,
                                 top_k, top_p)
        out = torch.cat([out, next_tok], dim=1)
    return out
