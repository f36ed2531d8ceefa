import os, sys, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29551")
import deepspeed_amd
from deepspeed_amd.models import MixtralForCausalLM, mixtral_mini

mode = sys.argv[1] if len(sys.argv) > 1 else "full"
torch.manual_seed(0)
cfg = mixtral_mini()
with torch.device("cuda"):
    model = MixtralForCausalLM(cfg)
model = model.to(torch.bfloat16)
print("built", flush=True)
if mode == "fwd_only":
    ids = torch.randint(0, cfg.vocab_size, (2, 512), device="cuda")
    with torch.no_grad():
        loss = model(ids, labels=ids)
    torch.cuda.synchronize()
    print("fwd ok", loss.item(), flush=True)
    sys.exit(0)
engine, _, _, _ = deepspeed_amd.initialize(model=model, config={
    "train_micro_batch_size_per_gpu": 2,
    "bf16": {"enabled": True},
    "zero_optimization": {"stage": 2},
    "optimizer": {"type": "AdamW", "params": {"lr": 1e-4}}})
ids = torch.randint(0, cfg.vocab_size, (2, 1024), device="cuda")
loss = engine(ids, labels=ids)
print("fwd ok", loss.item(), flush=True)
engine.backward(loss)
print("bwd ok", flush=True)
engine.step()
torch.cuda.synchronize()
print("step ok", flush=True)
