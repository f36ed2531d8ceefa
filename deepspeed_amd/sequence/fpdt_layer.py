"""FPDT-style chunked long-context attention (reference:
deepspeed/sequence/fpdt_layer.py — Fully Pipelined Distributed Transformer:
sequence chunked with online-softmax/LSE merging, chunks offloaded to host
between uses, :58 update_out_and_lse / :462 SequenceChunk).

This is the single-device core of FPDT: exact causal attention over
sequences far beyond HBM by processing Q in chunks and streaming KV chunks
(optionally parked in pinned host memory) through the online-softmax merge.
Composes with Ulysses (each SP rank runs this on its head shard after the
all-to-all). 288 GB of HBM3E pushes the need out to ~1M tokens, hence
chunking + host DRAM rather than smaller tiles.
"""

import math
from typing import List, Optional

import torch


class SequenceChunk:
    """A KV chunk that can live in pinned host memory between uses
    (reference fpdt_layer.py:462)."""

    def __init__(self, k: torch.Tensor, v: torch.Tensor, offload: bool):
        self.device = k.device
        self.offload = offload and k.is_cuda
        if self.offload:
            self.k = k.to("cpu", non_blocking=True).pin_memory() \
                if not k.is_pinned() else k.cpu()
            self.v = v.to("cpu", non_blocking=True).pin_memory()
        else:
            self.k, self.v = k, v

    def fetch(self):
        if self.offload:
            return (self.k.to(self.device, non_blocking=True),
                    self.v.to(self.device, non_blocking=True))
        return self.k, self.v


@torch.no_grad()
def chunked_prefill_attention(q: torch.Tensor, k: torch.Tensor,
                              v: torch.Tensor, chunk_size: int = 1024,
                              kv_offload: bool = False,
                              scale: Optional[float] = None) -> torch.Tensor:
    """Exact causal attention over [B, H, S, D] computed in sequence chunks
    with online log-sum-exp merging. Peak memory is O(chunk^2) scores
    instead of O(S^2); with ``kv_offload`` the KV tensors live in host DRAM
    between uses."""
    B, H, S, D = q.shape
    Hkv = k.shape[1]
    rep = H // Hkv
    scale = scale if scale is not None else 1.0 / math.sqrt(D)
    out = torch.empty_like(q)

    chunks: List[SequenceChunk] = []
    for ks in range(0, S, chunk_size):
        ke = min(ks + chunk_size, S)
        chunks.append(SequenceChunk(k[:, :, ks:ke], v[:, :, ks:ke],
                                    kv_offload))

    for qs in range(0, S, chunk_size):
        qe = min(qs + chunk_size, S)
        qc = q[:, :, qs:qe].float()
        m = torch.full((B, H, qe - qs, 1), -float("inf"), device=q.device)
        l = torch.zeros((B, H, qe - qs, 1), device=q.device)
        o = torch.zeros((B, H, qe - qs, D), device=q.device)
        for ci, ks in enumerate(range(0, qe, chunk_size)):
            ke = min(ks + chunk_size, S)
            kc, vc = chunks[ci].fetch()
            kc, vc = kc.float(), vc.float()
            if rep > 1:
                kc = kc.repeat_interleave(rep, dim=1)
                vc = vc.repeat_interleave(rep, dim=1)
            s = (qc @ kc.transpose(-1, -2)) * scale
            if ke > qs:  # diagonal chunk: causal mask
                qpos = torch.arange(qs, qe, device=q.device)[:, None]
                kpos = torch.arange(ks, ke, device=q.device)[None, :]
                s = s.masked_fill(kpos > qpos, -float("inf"))
            m_new = torch.maximum(m, s.amax(dim=-1, keepdim=True))
            alpha = torch.exp(m - m_new)
            p = torch.exp(s - m_new)
            l = l * alpha + p.sum(dim=-1, keepdim=True)
            o = o * alpha + p @ vc
            m = m_new
        out[:, :, qs:qe] = (o / l).to(q.dtype)
    return out
