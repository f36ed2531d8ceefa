import sys, os, math, time, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from deepspeed_amd.ops import _C
from deepspeed_amd.ops.attention import _t_last2_bhsd
B, S, H, Hkv, D = 8, 4096, 32, 8, 128
torch.manual_seed(0)
qh = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
kh = torch.randn(B, Hkv, S, D, device="cuda", dtype=torch.bfloat16)
vh = torch.randn(B, Hkv, S, D, device="cuda", dtype=torch.bfloat16)
doh = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
lse = (torch.randn(B, H, S, device="cuda") + 8).float()
delta = torch.randn(B, H, S, device="cuda").float()
scale = 1.0 / math.sqrt(D)
qt2, kt2, dot2 = _t_last2_bhsd(qh), _t_last2_bhsd(kh), _t_last2_bhsd(doh)
dk_o, dv_o = torch.empty_like(kh), torch.empty_like(vh)

def t(fn, n=10):
    for _ in range(3): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / n * 1e3

both = t(lambda: _C.flash_attn_bwd(qh, kh, vh, doh, qt2, kt2, dot2, lse, delta, scale, True))
dkdv = t(lambda: _C.flash_bwd_dkdv_dbg(qh, kh, vh, doh, qt2, dot2, lse, delta, dk_o, dv_o, scale, 0))
print(f"both: {both:7.2f} ms   dkdv-only: {dkdv:7.2f} ms   implied dq: {both-dkdv:7.2f} ms")
