"""Serving example: FastGen-style continuous batching with the paged KV
cache, SplitFuse token packing, and the paged flash-decode kernel.

    python examples/serve_continuous_batching.py          # offline demo
    python examples/serve_continuous_batching.py --http   # FastAPI server
"""
import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
import argparse

import torch

from deepspeed_amd.inference.ragged import (ContinuousBatcher, PagedKVCache,
                                            Request)
from deepspeed_amd.models import LlamaForCausalLM
from deepspeed_amd.models.llama import LlamaConfig


def build_model():
    cfg = LlamaConfig(vocab_size=32000, hidden_size=1024,
                      intermediate_size=2816, num_layers=8, num_heads=8,
                      num_kv_heads=2, max_seq_len=4096)  # head_dim=128
    torch.manual_seed(0)
    model = LlamaForCausalLM(cfg)
    if torch.cuda.is_available():
        model = model.cuda().bfloat16()
    return model.eval()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--http", action="store_true")
    args = ap.parse_args()
    model = build_model()
    dtype = torch.bfloat16 if torch.cuda.is_available() else torch.float32

    if args.http:
        import uvicorn
        from deepspeed_amd.inference.server import build_app
        app = build_app(model, max_slots=8, dtype=dtype)
        uvicorn.run(app, host="127.0.0.1", port=8000)
        return

    # offline: paged KV blocks + token-budget SplitFuse; on GPU the decode
    # steps run through the paged flash-decode HIP kernel
    batcher = ContinuousBatcher(model, max_slots=8, dtype=dtype,
                                cache_cls=PagedKVCache, token_budget=256)
    for i in range(4):
        prompt = torch.randint(0, 32000, (16 + 8 * i,))
        batcher.put(Request(uid=i, prompt=prompt, max_new_tokens=32,
                            do_sample=(i % 2 == 1), temperature=0.8,
                            top_p=0.95))
    for r in batcher.run_to_completion():
        print(f"request {r.uid}: {len(r.generated)} tokens ->",
              r.generated[:8], "...")


if __name__ == "__main__":
    main()
