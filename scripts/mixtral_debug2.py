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
    # inline the rest of MOELayer.forward with syncs
    dispatched = tokens.new_zeros((E * C, d_model))
    dispatched.index_add_(0, dest, tokens[tok])
    torch.cuda.synchronize(); print("dispatch ok", flush=True)
    from deepspeed_amd.moe.sharded_moe import _AllToAll
    dispatched = _AllToAll.apply(self.ep_group, dispatched)
    torch.cuda.synchronize(); print("a2a ok", flush=True)
    dispatched = dispatched.reshape(self.ep_size, self.num_local_experts,
                                    C, d_model)
    chunks = dispatched.transpose(0, 1).reshape(
        self.num_local_experts, self.ep_size * C, d_model)
    torch.cuda.synchronize(); print("chunks ok", chunks.shape,
                                    chunks.is_contiguous(), flush=True)
    from deepspeed_amd.moe.experts import FusedExperts
    if isinstance(self.experts, FusedExperts):
        g = torch.bmm(chunks, self.experts.w_gate.transpose(1, 2))
        torch.cuda.synchronize(); print("bmm gate ok", flush=True)
        u = torch.bmm(chunks, self.experts.w_up.transpose(1, 2))
        torch.cuda.synchronize(); print("bmm up ok", flush=True)
        from deepspeed_amd.ops import swiglu as _sw
        h = _sw(g, u)
        torch.cuda.synchronize(); print("swiglu ok", flush=True)
        expert_out = torch.bmm(h, self.experts.w_down.transpose(1, 2))
        torch.cuda.synchronize(); print("bmm down ok", flush=True)
    else:
        outs = [e(chunks[i]) for i, e in
                enumerate(self.experts.local_experts)]
        expert_out = torch.stack(outs, dim=0)
        torch.cuda.synchronize(); print("loop experts ok", flush=True)
    expert_out = expert_out.reshape(self.num_local_experts, self.ep_size,
                                    C, d_model).transpose(0, 1)
    expert_out = _AllToAll.apply(self.ep_group,
                                 expert_out.reshape(E * C, d_model))
    torch.cuda.synchronize(); print("a2a2 ok", flush=True)
    gathered = expert_out[dest]
    weighted = gathered * route["gate"].unsqueeze(1).to(gathered.dtype)
    out = tokens.new_zeros((T, d_model))
    out.index_add_(0, tok, weighted)
    torch.cuda.synchronize(); print("combine ok", flush=True)
    self.l_aux = laux
    self.exp_counts = route["exp_counts"]
    return out.reshape(x.shape)
sharded_moe.MOELayer.forward = dbg_forward

ids = torch.randint(0, cfg.vocab_size, (mb, S), device="cuda")
with torch.no_grad():
    loss = model(ids, labels=ids)
torch.cuda.synchronize()
print("fwd ok", loss.item(), flush=True)
