"""Minimal training example: Llama-3-8B ZeRO-3 bf16 on synthetic data.

Launch on one MI355X node:
    python -m deepspeed_amd.launcher.runner --num_gpus 8 examples/train_llama.py
"""

import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
import torch

import deepspeed_amd
from deepspeed_amd.models import LlamaForCausalLM, llama3_8b

CONFIG = {
    "train_micro_batch_size_per_gpu": 8,
    "gradient_accumulation_steps": 1,
    "bf16": {"enabled": True},
    "gradient_clipping": 1.0,
    "zero_optimization": {"stage": 3, "overlap_comm": True},
    "optimizer": {"type": "AdamW",
                  "params": {"lr": 1e-4, "betas": [0.9, 0.95],
                             "weight_decay": 0.1}},
    "scheduler": {"type": "WarmupLR",
                  "params": {"warmup_num_steps": 100}},
    "monitor_config": {"enabled": True,
                       "csv_monitor": {"enabled": True,
                                       "output_path": "runs"}},
}


def main():
    torch.manual_seed(42)
    model = LlamaForCausalLM(llama3_8b())
    engine, _, _, _ = deepspeed_amd.initialize(model=model, config=CONFIG)

    seq = 4096
    for step in range(1000):
        ids = torch.randint(0, model.cfg.vocab_size,
                            (CONFIG["train_micro_batch_size_per_gpu"], seq + 1),
                            device=engine.device)
        loss = engine(ids[:, :-1], labels=ids[:, 1:])
        engine.backward(loss)
        engine.step()
        if step % 10 == 0 and engine.global_rank == 0:
            print(f"step {step} loss {loss.item():.4f} "
                  f"lr {engine.get_lr()[0]:.2e}")
        if step % 200 == 199:
            engine.save_checkpoint("checkpoints")


if __name__ == "__main__":
    main()
