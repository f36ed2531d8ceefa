import sys, os, math, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from deepspeed_amd.ops import _C
from deepspeed_amd.ops.attention import flash_attn_bwd
B, S, H, Hkv, D = 8, 4096, 32, 8, 128
torch.manual_seed(0)
q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
k = torch.randn(B, Hkv, S, D, device="cuda", dtype=torch.bfloat16)
v = torch.randn(B, Hkv, S, D, device="cuda", dtype=torch.bfloat16)
o = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
do = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
lse = torch.randn(B, H, S, device="cuda", dtype=torch.float32) + 8
for _ in range(3):
    dq, dk, dv = flash_attn_bwd(q, k, v, o, do, lse, True)
torch.cuda.synchronize()
