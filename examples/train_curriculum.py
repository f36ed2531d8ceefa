"""Curriculum-learning example: data_efficiency config drives the
difficulty schedule; the engine dataloader samples only sequences at or
below the current difficulty (difficulty = sample length by default)."""
import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
import torch

import deepspeed_amd
from deepspeed_amd.models import GPT2ForCausalLM, gpt2_tiny


class VarLenDataset(torch.utils.data.Dataset):
    def __len__(self):
        return 512

    def __getitem__(self, i):
        n = 8 + (i % 25)  # lengths 8..32
        g = torch.Generator().manual_seed(i)
        return torch.randint(0, 128, (n,), generator=g)


def collate(batch):
    L = max(len(b) for b in batch)
    ids = torch.zeros(len(batch), L, dtype=torch.long)
    for i, b in enumerate(batch):
        ids[i, :len(b)] = b
    return ids


def main():
    model = GPT2ForCausalLM(gpt2_tiny())
    engine, _, _, _ = deepspeed_amd.initialize(model=model, config={
        "train_micro_batch_size_per_gpu": 4,
        "optimizer": {"type": "AdamW", "params": {"lr": 3e-4}},
        "data_efficiency": {
            "enabled": True,
            "curriculum_learning": {
                "enabled": True,
                "curriculum_type": "fixed_linear",
                "min_difficulty": 10, "max_difficulty": 32,
                "schedule_config": {"total_curriculum_step": 50,
                                    "difficulty_step": 2}}},
        "monitor_config": {"enabled": True,
                           "csv_monitor": {"enabled": True,
                                           "output_path": "/tmp/curri"}},
    })
    loader = engine.deepspeed_io(VarLenDataset(), collate_fn=collate)
    for step, ids in enumerate(loader):
        ids = ids.to(engine.device)
        loss = engine(ids, labels=ids)
        engine.backward(loss)
        engine.step()
        if step % 10 == 0:
            d = engine.curriculum_scheduler.get_current_difficulty()
            print(f"step {step:3d} loss {loss.item():.3f} "
                  f"difficulty<= {d} seq_len {ids.shape[1]}")
        if step >= 60:
            break


if __name__ == "__main__":
    main()
