"""Minimal inference example: TP-sharded KV-cached generation.

    python -m deepspeed_amd.launcher.runner --num_gpus 2 examples/generate.py
"""

import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
import torch

import deepspeed_amd
from deepspeed_amd.models import LlamaForCausalLM, llama_mini


def main():
    torch.manual_seed(0)
    model = LlamaForCausalLM(llama_mini())
    import os
    tp = int(os.environ.get("WORLD_SIZE", 1))
    engine = deepspeed_amd.init_inference(
        model, dtype=torch.bfloat16 if torch.cuda.is_available()
        else torch.float32,
        tensor_parallel={"tp_size": tp})

    prompt = torch.randint(0, model.cfg.vocab_size, (2, 16))
    out = engine.generate(prompt, max_new_tokens=32, do_sample=True,
                          temperature=0.8, top_k=50)
    if engine.tp_rank == 0:
        print("generated token ids:", out.tolist())


if __name__ == "__main__":
    main()
