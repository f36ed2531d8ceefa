import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import math, torch
from deepspeed_amd.ops import _C
from deepspeed_amd.ops.attention import sdpa_reference
torch.manual_seed(0)
B, S, H, Hkv, D = 1, 256, 4, 2, 128
q = torch.randn(B, S, H, D, device="cuda", dtype=torch.bfloat16)
k = torch.randn(B, S, Hkv, D, device="cuda", dtype=torch.bfloat16)
v = torch.randn(B, S, Hkv, D, device="cuda", dtype=torch.bfloat16)
vt = v.permute(0, 2, 3, 1).contiguous()
ref = sdpa_reference(q, k, v, causal=True).float()
scale = 1.0 / math.sqrt(D)
for var in (0, 1, 2, 3, 4):
    o = _C.flash_attn_fwd_dbg(q.contiguous(), k.contiguous(), vt, scale, var).float()
    err = (o - ref).abs()
    rel = (err.max() / ref.abs().max()).item()
    # per-wave error: which q-row blocks are wrong?
    per32 = err.view(B, S // 32, 32, H, D).amax(dim=(0, 2, 3, 4))
    print(f"var={var} maxrel={rel:.4f} per-32-rows={[round(x,3) for x in per32.tolist()]}")
