"""Flash attention A/B bench on the bench shape (Llama-3-8B mb=8 s=4096).

Times: our fwd kernel (all dbg variants), SDPA fwd, SDPA bwd, our bwd
kernels. Run on the GPU box:
    python scripts/flash_bench.py [B S H Hkv]
"""
import sys, os, time, math
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.nn.functional as F

B, S, H, Hkv, D = 8, 4096, 32, 8, 128
if len(sys.argv) > 4:
    B, S, H, Hkv = map(int, sys.argv[1:5])

from deepspeed_amd.ops import _C

torch.manual_seed(0)
dev = "cuda"
q = torch.randn(B, S, H, D, device=dev, dtype=torch.bfloat16)
k = torch.randn(B, S, Hkv, D, device=dev, dtype=torch.bfloat16)
v = torch.randn(B, S, Hkv, D, device=dev, dtype=torch.bfloat16)
vt = v.permute(0, 2, 3, 1).contiguous()
scale = 1.0 / math.sqrt(D)
fwd_flops = 4 * B * H * S * S * D / 2  # causal


def time_fn(fn, n=10, warm=3):
    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n


# ---- fwd variants
for var in sorted(set([3, 0, 1, 2, 4] if os.environ.get("ABLATE") else [3, 4])):
    try:
        t = time_fn(lambda: _C.flash_attn_fwd_dbg(q, k, vt, scale, var))
        print(f"fwd dbg var={var}: {t*1e3:7.2f} ms  {fwd_flops/t/1e12:6.0f} TF")
    except Exception as e:
        print(f"fwd dbg var={var}: FAILED {e}")

o_lse = _C.flash_attn_fwd_lse(q, k, vt, scale, True)
o, lse = o_lse
t = time_fn(lambda: _C.flash_attn_fwd_lse(q, k, vt, scale, True))
print(f"fwd (+lse):     {t*1e3:7.2f} ms  {fwd_flops/t/1e12:6.0f} TF")

# ---- SDPA fwd / bwd
qt_, kt_, vt_ = (x.transpose(1, 2).contiguous() for x in (q, k, v))
with torch.no_grad():
    t = time_fn(lambda: F.scaled_dot_product_attention(
        qt_, kt_, vt_, is_causal=True, enable_gqa=True))
print(f"SDPA fwd:       {t*1e3:7.2f} ms  {fwd_flops/t/1e12:6.0f} TF")

qg = qt_.detach().requires_grad_(True)
kg = kt_.detach().requires_grad_(True)
vg = vt_.detach().requires_grad_(True)
do_b = torch.randn_like(qt_)


def sdpa_fb():
    out = F.scaled_dot_product_attention(qg, kg, vg, is_causal=True,
                                         enable_gqa=True)
    torch.autograd.backward(out, do_b)
    qg.grad = kg.grad = vg.grad = None


t_fb = time_fn(sdpa_fb)
print(f"SDPA fwd+bwd:   {t_fb*1e3:7.2f} ms  ({fwd_flops*3.5/t_fb/1e12:6.0f} TF eff)")

# ---- our bwd kernels (BHSD layouts + transposes made by wrapper)
qh, kh, vh = (x.transpose(1, 2).contiguous() for x in (q, k, v))
oh = o.transpose(1, 2).contiguous()
doh = do_b.contiguous()
from deepspeed_amd.ops.attention import flash_attn_bwd, _t_last2_bhsd
t = time_fn(lambda: flash_attn_bwd(qh, kh, vh, oh, doh, lse, True), n=5)
bwd_flops = fwd_flops * 2.5
print(f"our bwd (wrap): {t*1e3:7.2f} ms  {bwd_flops/t/1e12:6.0f} TF")

# raw kernel split (pre-transposed inputs)
delta = (doh.float() * oh.float()).sum(-1)
qt2, kt2, dot2 = _t_last2_bhsd(qh), _t_last2_bhsd(kh), _t_last2_bhsd(doh)
t = time_fn(lambda: _C.flash_attn_bwd(qh, kh, vh, doh, qt2, kt2, dot2,
                                      lse.float(), delta, scale, True), n=5)
print(f"bwd kernels:    {t*1e3:7.2f} ms  {bwd_flops/t/1e12:6.0f} TF")
dk_o = torch.empty_like(kh)
dv_o = torch.empty_like(vh)
t = time_fn(lambda: _C.flash_bwd_dkdv_dbg(qh, kh, vh, doh, qt2, dot2,
                                          lse.float(), delta, dk_o, dv_o,
                                          scale, 0), n=10)
print(f"dkdv only:      {t*1e3:7.2f} ms")
t = time_fn(lambda: _C.flash_bwd_dkdv_dbg(qh, kh, vh, doh, qt2, dot2,
                                          lse.float(), delta, dk_o, dv_o,
                                          scale, 0), n=30)
print(f"dkdv only n30:  {t*1e3:7.2f} ms")
t = time_fn(lambda: _t_last2_bhsd(qh))
print(f"transpose k:    {t*1e3:7.2f} ms  ({qh.numel()*4/t/1e12:5.2f} TB/s)")
t = time_fn(lambda: qh.transpose(-1, -2).contiguous())
print(f"transpose torch:{t*1e3:7.2f} ms  ({qh.numel()*4/t/1e12:5.2f} TB/s)")
print("done")
