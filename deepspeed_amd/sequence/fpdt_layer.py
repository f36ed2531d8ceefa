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

    pipe = _PrefetchPipeline(chunks) if kv_offload else None
    for qs in range(0, S, chunk_size):
        qe = min(qs + chunk_size, S)
        qc = q[:, :, qs:qe].float()
        m = torch.full((B, H, qe - qs, 1), -float("inf"), device=q.device)
        l = torch.zeros((B, H, qe - qs, 1), device=q.device)
        o = torch.zeros((B, H, qe - qs, D), device=q.device)
        for ci, ks in enumerate(range(0, qe, chunk_size)):
            ke = min(ks + chunk_size, S)
            kc, vc = pipe.get(ci) if pipe is not None else chunks[ci].fetch()
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


class _PrefetchPipeline:
    """Double-buffered H2D prefetch of offloaded KV chunks on a side
    stream (the reference's offload pipeline, fpdt_layer.py:510
    _FPDTGPUOffloadingAttentionImpl_): while chunk i is consumed by
    compute, chunk i+1 streams host->device."""

    def __init__(self, chunks: List[SequenceChunk]):
        self.chunks = chunks
        self.stream = torch.cuda.Stream() if torch.cuda.is_available() \
            else None
        self._staged = {}

    def _issue(self, idx):
        if idx >= len(self.chunks) or idx in self._staged:
            return
        if self.stream is not None:
            with torch.cuda.stream(self.stream):
                kv = self.chunks[idx].fetch()
                ev = torch.cuda.Event()
                ev.record(self.stream)
                self._staged[idx] = (kv, ev)
        else:
            self._staged[idx] = (self.chunks[idx].fetch(), None)

    def get(self, idx):
        self._issue(idx)
        self._issue(idx + 1)  # prefetch ahead
        kv, ev = self._staged.pop(idx)
        if ev is not None:
            torch.cuda.current_stream().wait_event(ev)
        return kv


class _FPDTAttnFunc(torch.autograd.Function):
    """Differentiable chunked causal attention with online-LSE forward and
    chunk-recompute backward — exact flash semantics at O(chunk^2) peak
    memory, so FPDT serves TRAINING-length contexts, not just prefill."""

    @staticmethod
    def forward(ctx, q, k, v, chunk_size, scale):
        B, H, S, D = q.shape
        rep = H // k.shape[1]
        out = torch.empty_like(q)
        lse = torch.empty(B, H, S, device=q.device, dtype=torch.float32)
        with torch.no_grad():
            for qs in range(0, S, chunk_size):
                qe = min(qs + chunk_size, S)
                qc = q[:, :, qs:qe].float()
                m = torch.full((B, H, qe - qs, 1), -float("inf"),
                               device=q.device)
                l = torch.zeros((B, H, qe - qs, 1), device=q.device)
                o = torch.zeros((B, H, qe - qs, D), device=q.device)
                for ks in range(0, qe, chunk_size):
                    ke = min(ks + chunk_size, S)
                    kc = k[:, :, ks:ke].float()
                    vc = v[:, :, ks:ke].float()
                    if rep > 1:
                        kc = kc.repeat_interleave(rep, dim=1)
                        vc = vc.repeat_interleave(rep, dim=1)
                    s = (qc @ kc.transpose(-1, -2)) * scale
                    if ke > qs:
                        qpos = torch.arange(qs, qe, device=q.device)[:, None]
                        kpos = torch.arange(ks, ke, device=q.device)[None, :]
                        s = s.masked_fill(kpos > qpos, -float("inf"))
                    m_new = torch.maximum(m, s.amax(-1, keepdim=True))
                    alpha = torch.exp(m - m_new)
                    l = l * alpha + torch.exp(s - m_new).sum(-1, keepdim=True)
                    o = o * alpha + torch.exp(s - m_new) @ vc
                    m = m_new
                out[:, :, qs:qe] = (o / l).to(q.dtype)
                lse[:, :, qs:qe] = (m + l.log()).squeeze(-1)
        ctx.save_for_backward(q, k, v, out, lse)
        ctx.chunk_size, ctx.scale = chunk_size, scale
        return out

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse = ctx.saved_tensors
        C, scale = ctx.chunk_size, ctx.scale
        B, H, S, D = q.shape
        Hkv = k.shape[1]
        rep = H // Hkv
        dq = torch.zeros_like(q, dtype=torch.float32)
        dkh = torch.zeros(B, H, S, D, device=q.device, dtype=torch.float32)
        dvh = torch.zeros_like(dkh)
        delta = (do.float() * o.float()).sum(-1)  # [B,H,S]
        for qs in range(0, S, C):
            qe = min(qs + C, S)
            qc = q[:, :, qs:qe].float()
            doc = do[:, :, qs:qe].float()
            lsec = lse[:, :, qs:qe].unsqueeze(-1)
            dlc = delta[:, :, qs:qe].unsqueeze(-1)
            for ks in range(0, qe, C):
                ke = min(ks + C, S)
                kc = k[:, :, ks:ke].float()
                vc = v[:, :, ks:ke].float()
                if rep > 1:
                    kc = kc.repeat_interleave(rep, dim=1)
                    vc = vc.repeat_interleave(rep, dim=1)
                s = (qc @ kc.transpose(-1, -2)) * scale
                if ke > qs:
                    qpos = torch.arange(qs, qe, device=q.device)[:, None]
                    kpos = torch.arange(ks, ke, device=q.device)[None, :]
                    s = s.masked_fill(kpos > qpos, -float("inf"))
                p = torch.exp(s - lsec)
                dp = doc @ vc.transpose(-1, -2)
                ds = p * (dp - dlc) * scale
                dq[:, :, qs:qe] += ds @ kc
                dkh[:, :, ks:ke] += ds.transpose(-1, -2) @ qc
                dvh[:, :, ks:ke] += p.transpose(-1, -2) @ doc
        if rep > 1:
            dkh = dkh.view(B, Hkv, rep, S, D).sum(2)
            dvh = dvh.view(B, Hkv, rep, S, D).sum(2)
        return (dq.to(q.dtype), dkh.to(k.dtype), dvh.to(v.dtype),
                None, None)


def fpdt_attention(q, k, v, chunk_size: int = 1024,
                   scale: Optional[float] = None) -> torch.Tensor:
    """Differentiable FPDT chunked causal attention ([B,H,S,D], GQA)."""
    scale = scale if scale is not None else 1.0 / math.sqrt(q.size(-1))
    return _FPDTAttnFunc.apply(q, k, v, chunk_size, scale)


class FPDTAttention(torch.nn.Module):
    """Drop-in local-attention module for Ulysses' DistributedAttention:
    consumes [b, s, H, d] (BSHD), runs chunked exact causal attention —
    the long-context per-rank kernel of the Fully Pipelined Distributed
    Transformer (composes SP a2a + sequence chunking)."""

    def __init__(self, chunk_size: int = 1024, kv_offload: bool = False):
        super().__init__()
        self.chunk_size = chunk_size
        self.kv_offload = kv_offload

    def forward(self, q, k, v):
        qh, kh, vh = (t.transpose(1, 2) for t in (q, k, v))
        if torch.is_grad_enabled() and (q.requires_grad or k.requires_grad):
            o = fpdt_attention(qh, kh, vh, self.chunk_size)
        else:
            o = chunked_prefill_attention(qh, kh, vh, self.chunk_size,
                                          kv_offload=self.kv_offload)
        return o.transpose(1, 2)
