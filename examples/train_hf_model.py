"""Train any HuggingFace causal-LM under the engine (the reference's most
common user path). Shows ZeRO-3 with the ZeRO++ knobs (hpZ/qwZ/qgZ) and
how to switch the optimizer state to NVMe (ZeRO-Infinity).

Launch on one MI355X node:
    python -m deepspeed_amd.launcher.runner --num_gpus 8 \
        examples/train_hf_model.py
Multi-node with elastic restarts:
    python -m deepspeed_amd.launcher.runner --hostfile hosts \
        --max_restarts 3 examples/train_hf_model.py
"""

import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
import torch

import deepspeed_amd
from deepspeed_amd.runtime.zero.partition import Init

CONFIG = {
    "train_micro_batch_size_per_gpu": 4,
    "bf16": {"enabled": True},
    "gradient_clipping": 1.0,
    "zero_optimization": {
        "stage": 3,
        "overlap_comm": True,
        # ZeRO++ (multi-node): uncomment to localize weight gathers to a
        # node of 8 ranks and quantize the cross-node traffic
        # "zero_hpz_partition_size": 8,
        # "zero_quantized_weights": True,
        # "zero_quantized_gradients": True,
        # ZeRO-Infinity: put optimizer state on NVMe instead
        # "offload_optimizer": {"device": "nvme",
        #                       "nvme_path": "/local_nvme/zero"},
        # "sub_group_size": 1 << 26,   # swap chunk (elements)
    },
    "optimizer": {"type": "AdamW", "params": {"lr": 1e-4}},
}


def main():
    from transformers import LlamaConfig, LlamaForCausalLM
    torch.manual_seed(42)
    # zero.Init casts parameters to bf16 as they are constructed, halving
    # the transient host footprint for big models
    with Init(dtype=torch.bfloat16):
        model = LlamaForCausalLM(LlamaConfig(
            num_hidden_layers=4, hidden_size=512, intermediate_size=1408,
            num_attention_heads=8, num_key_value_heads=4, vocab_size=32000))
    engine, _, _, _ = deepspeed_amd.initialize(model=model, config=CONFIG)
    for step in range(10):
        ids = torch.randint(0, 32000, (4, 256), device=engine.device)
        loss = engine(ids, labels=ids).loss
        engine.backward(loss)
        engine.step()
        if engine.global_rank == 0:
            print(f"step {step}: loss {loss.item():.4f}")


if __name__ == "__main__":
    main()
