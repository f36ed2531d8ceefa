import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import math, torch
from deepspeed_amd.ops import _C
from deepspeed_amd.ops.attention import sdpa_reference
torch.manual_seed(0)

# shapes: (B, S, H, Hkv) — S%32==0 required; include S%64!=0 and MHA
for (B, S, H, Hkv) in [(1, 256, 4, 2), (2, 160, 4, 4), (1, 4096, 8, 2),
                       (3, 96, 2, 1)]:
    D = 128
    q = torch.randn(B, S, H, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, S, Hkv, D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, S, Hkv, D, device="cuda", dtype=torch.bfloat16)
    vt = v.permute(0, 2, 3, 1).contiguous()
    scale = 1.0 / math.sqrt(D)
    ref = sdpa_reference(q, k, v, causal=True).float()
    for var in (0, 3):
        o = _C.flash_attn_fwd_dbg(q.contiguous(), k.contiguous(), vt, scale, var).float()
        err = (o - ref).abs()
        rel = (err.max() / ref.abs().max()).item()
        per32 = err.view(B, S // 32, 32, H, D).amax(dim=(0, 2, 3, 4))
        flag = "OK " if rel < 0.02 else "BAD"
        print(f"{flag} B{B} S{S} H{H}/{Hkv} var={var} maxrel={rel:.4f} "
              f"per-32-rows={[round(x,3) for x in per32.tolist()[:8]]}")
    # non-causal via main entry
    refnc = sdpa_reference(q, k, v, causal=False).float()
    onc = _C.flash_attn_fwd(q.contiguous(), k.contiguous(), vt, scale, False).float()
    rel = ((onc - refnc).abs().max() / refnc.abs().max()).item()
    print(f"{'OK ' if rel < 0.02 else 'BAD'} B{B} S{S} noncausal maxrel={rel:.4f}")
    # lse path
    o2, lse = _C.flash_attn_fwd_lse(q.contiguous(), k.contiguous(), vt, scale, True)
    qt, kt2, vt2 = (t.transpose(1, 2).float() for t in (q, k, v))
    rep = H // Hkv
    if rep > 1:
        kt2 = kt2.repeat_interleave(rep, dim=1)
    sc = (qt @ kt2.transpose(-1, -2)) * scale
    mask = torch.ones(S, S, device="cuda", dtype=torch.bool).tril()
    sc = sc.masked_fill(~mask, float("-inf"))
    lse_ref = torch.logsumexp(sc, dim=-1)  # [B,H,S]
    rel = ((lse - lse_ref).abs().max()).item()
    print(f"{'OK ' if rel < 0.02 else 'BAD'} B{B} S{S} lse maxerr={rel:.4f}")
