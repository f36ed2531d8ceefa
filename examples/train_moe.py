"""MoE training example: Mixtral-architecture model with expert
parallelism (EP over the xGMI mesh when launched with
`python -m deepspeed_amd.launcher.runner --num_gpus 8 examples/train_moe.py`),
grouped expert GEMMs, expert-DP-aware ZeRO-2, and per-EP-rank expert
checkpoints."""
import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
import torch

import deepspeed_amd
from deepspeed_amd.models import MixtralForCausalLM, mixtral_tiny


def main():
    world = 1
    import os
    world = int(os.environ.get("WORLD_SIZE", 1))
    cfg = mixtral_tiny(ep_size=min(world, 4), num_experts=4)
    torch.manual_seed(7)
    model = MixtralForCausalLM(cfg)
    engine, _, _, _ = deepspeed_amd.initialize(model=model, config={
        "train_micro_batch_size_per_gpu": 4,
        "bf16": {"enabled": torch.cuda.is_available()},
        "zero_optimization": {"stage": 2, "overlap_comm": True},
        "optimizer": {"type": "AdamW", "params": {"lr": 3e-4}}})
    torch.manual_seed(100 + engine.global_rank)
    for step in range(20):
        ids = torch.randint(0, cfg.vocab_size, (4, 64),
                            device=engine.device)
        loss = engine(ids, labels=ids)
        engine.backward(loss)
        engine.step()
        if step % 5 == 0 and engine.global_rank == 0:
            print(f"step {step:2d} loss {loss.item():.3f} "
                  f"aux {float(engine.module.aux_loss()):.4f}")
    engine.save_checkpoint("/tmp/moe_ckpt")  # incl. per-EP expert shards


if __name__ == "__main__":
    main()
