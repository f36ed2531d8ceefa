import os, sys, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29553")
from deepspeed_amd.models import MixtralForCausalLM, mixtral_mini
from deepspeed_amd.moe import sharded_moe

mb = int(sys.argv[1]) if len(sys.argv) > 1 else 8
S = int(sys.argv[2]) if len(sys.argv) > 2 else 2048
torch.manual_seed(0)
cfg = mixtral_mini()
with torch.device("cuda"):
    model = MixtralForCausalLM(cfg)
model = model.to(torch.bfloat16)
print("built", flush=True)

# sync + validate inside the MoE layer
orig = sharded_moe.MOELayer.forward
def dbg_forward(self, x, *a):
    torch.cuda.synchronize(); print("moe in", x.shape, flush=True)
    d_model = x.shape[-1]
    tokens = x.reshape(-1, d_model)
    laux, route = self.gate(tokens)
    torch.cuda.synchronize(); print("gate ok", flush=True)
    T = tokens.shape[0]; C = route["capacity"]; E = self.num_experts
    dest, tok = route["dest"], route["token"]
    print("T", T, "C", C, "E", E,
          "dest[min,max]", int(dest.min()), int(dest.max()),
          "tok[min,max]", int(tok.min()), int(tok.max()),
          "gate[min,max]", float(route["gate"].min()),
          float(route["gate"].max()), flush=True)
    assert int(dest.max()) < E * C and int(dest.min()) >= 0
    assert int(tok.max()) < T and int(tok.min()) >= 0
    return orig(self, x, *a)
sharded_moe.MOELayer.forward = dbg_forward

ids = torch.randint(0, cfg.vocab_size, (mb, S), device="cuda")
with torch.no_grad():
    loss = model(ids, labels=ids)
torch.cuda.synchronize()
print("fwd ok", loss.item(), flush=True)
