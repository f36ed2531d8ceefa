"""Tile-level reference of the CDNA4 flash-attention backward
(docs/flash_bwd_design.md). This mirrors the HIP kernels' exact loop
structure and math (32-row kv/q tiles, recompute from lse, delta trick,
GQA head-group accumulation) in torch ops, so the algebra is CPU-testable
against autograd before any GPU run; the HIP kernels must match this
tile-for-tile."""

import math

import torch


@torch.no_grad()
def flash_bwd_reference(q, k, v, o, do, lse, causal=True, scale=None,
                        tile=32):
    """q,o,do: [B,H,S,D]; k,v: [B,Hkv,S,D]; lse: [B,H,S].
    Returns (dq, dk, dv) with dk/dv in [B,Hkv,S,D] (summed over the
    query-head group)."""
    B, H, S, D = q.shape
    Hkv = k.shape[1]
    G = H // Hkv
    scale = scale if scale is not None else 1.0 / math.sqrt(D)
    qf, kf, vf, of, dof = (t.float() for t in (q, k, v, o, do))
    delta = (dof * of).sum(-1)                      # [B,H,S]

    dq = torch.zeros_like(qf)
    dk = torch.zeros_like(kf)
    dv = torch.zeros_like(vf)

    nt = (S + tile - 1) // tile
    for b in range(B):
        for h in range(H):
            hkv = h // G
            for j in range(nt):                     # kv tile (kernel 1 grid)
                ks, ke = j * tile, min((j + 1) * tile, S)
                Kt = kf[b, hkv, ks:ke]              # [t, D]
                Vt = vf[b, hkv, ks:ke]
                i0 = j if causal else 0
                for i in range(i0 if causal else 0, nt):
                    qs, qe = i * tile, min((i + 1) * tile, S)
                    Qt = qf[b, h, qs:qe]
                    dOt = dof[b, h, qs:qe]
                    st = (Kt @ Qt.T) * scale        # S^T [kv, q]
                    if causal:
                        kvpos = torch.arange(ks, ke)[:, None]
                        qpos = torch.arange(qs, qe)[None, :]
                        st = st.masked_fill(kvpos > qpos, -float("inf"))
                    pt = torch.exp(st - lse[b, h, qs:qe][None, :])
                    dpt = Vt @ dOt.T                # dP^T [kv, q]
                    dst = pt * (dpt - delta[b, h, qs:qe][None, :]) * scale
                    dv[b, hkv, ks:ke] += pt @ dOt
                    dk[b, hkv, ks:ke] += dst @ Qt
                    dq[b, h, qs:qe] += dst.T @ Kt   # kernel 2 recomputes this
    return dq.to(q.dtype), dk.to(k.dtype), dv.to(v.dtype)
