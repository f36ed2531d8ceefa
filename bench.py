"""Flagship benchmark: Llama-3-8B ZeRO training step, tokens/sec whole job.

Contract (driver): `python bench.py --gpus N --steps K --warmup W`.
For N>1 the driver launches via torch.distributed.run (one rank per GPU,
RCCL over xGMI); this script reads RANK/LOCAL_RANK/WORLD_SIZE/MASTER_* from
the environment. Rank 0 prints exactly ONE JSON line.

Metric per BASELINE.json: tokens/sec (whole node) for Llama-3-8B ZeRO bf16
on synthetic data of the benchmark's shape with random-init weights.
"""

import argparse
import json
import os
import time

import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=8)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--seq-len", type=int, default=4096)
    p.add_argument("--micro-batch", type=int, default=8)
    p.add_argument("--gas", type=int, default=1)
    p.add_argument("--zero-stage", type=int,
                   default=int(os.environ.get("BENCH_ZERO_STAGE", 3)))
    p.add_argument("--model", type=str, default="llama3-8b",
                   choices=["llama3-8b", "llama3-70b", "llama-mini", "tiny",
                            "phi3-mini", "mixtral-8x7b", "mixtral-mini"])
    p.add_argument("--offload", action="store_true",
                   help="ZeRO-Offload optimizer states to host DRAM")
    p.add_argument("--activation-checkpointing", action="store_true")
    p.add_argument("--local_rank", type=int, default=-1)
    return p.parse_args()


def build_model(name, world_size=1):
    """Construct the model directly on the GPU in bf16 (random init on
    device): building 8B on host fp32 took ~3 min of the round-1 bench
    lease; on-device construction is seconds."""
    torch.manual_seed(42)
    dev = torch.device("cuda") if torch.cuda.is_available() else         torch.device("cpu")
    if name.startswith("mixtral"):
        from deepspeed_amd.models import (MixtralForCausalLM, mixtral_8x7b,
                                          mixtral_mini)
        f = mixtral_8x7b if name == "mixtral-8x7b" else mixtral_mini
        cfg = f(ep_size=min(world_size, 8))
        with dev:
            model = MixtralForCausalLM(cfg)
        if dev.type == "cuda":
            model = model.to(torch.bfloat16)
        return model, cfg
    from deepspeed_amd.models import (LlamaForCausalLM, llama3_8b, llama3_70b,
                                      llama_mini, llama_tiny, phi3_mini)
    cfg = {"llama3-8b": llama3_8b, "llama3-70b": llama3_70b,
           "llama-mini": llama_mini, "tiny": llama_tiny,
           "phi3-mini": phi3_mini}[name]()
    with dev:
        model = LlamaForCausalLM(cfg)
    if dev.type == "cuda":
        model = model.to(torch.bfloat16)
    return model, cfg


def main():
    args = parse_args()
    world_size = int(os.environ.get("WORLD_SIZE", 1))
    rank = int(os.environ.get("RANK", 0))
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29517")

    import deepspeed_amd
    from deepspeed_amd import comm as dist

    use_gpu = torch.cuda.is_available()
    if use_gpu:
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", 0)))

    def log(msg):
        print(f"[bench rank{rank}] {msg}", file=__import__("sys").stderr,
              flush=True)

    log("building model...")
    t_build = time.time()
    model, cfg = build_model(args.model, world_size)
    n_params = model.num_parameters()
    log(f"model built in {time.time()-t_build:.1f}s ({n_params/1e9:.2f}B)")

    ds_config = {
        "train_micro_batch_size_per_gpu": args.micro_batch,
        "gradient_accumulation_steps": args.gas,
        "bf16": {"enabled": True},
        "gradient_clipping": 1.0,
        "zero_optimization": {
            "stage": args.zero_stage,
            "overlap_comm": True,
            "offload_optimizer": {"device": "cpu" if args.offload else "none"},
        },
        "optimizer": {"type": "AdamW",
                      "params": {"lr": 1e-4, "betas": [0.9, 0.95],
                                 "weight_decay": 0.1}},
    }

    if args.activation_checkpointing:
        model.model.gradient_checkpointing_enable()

    t_init = time.time()
    engine, _, _, _ = deepspeed_amd.initialize(model=model, config=ds_config)
    log(f"engine initialized in {time.time()-t_init:.1f}s")
    device = engine.device

    torch.manual_seed(1234 + rank)
    seq = args.seq_len
    data = [torch.randint(0, cfg.vocab_size, (args.micro_batch, seq + 1),
                          device=device) for _ in range(4)]

    def step(i):
        for _ in range(args.gas):
            batch = data[i % len(data)]
            loss = engine(batch[:, :-1], labels=batch[:, 1:])
            engine.backward(loss)
        engine.step()
        return loss

    for i in range(args.warmup):
        step(i)

    dist.barrier()
    if use_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(args.steps):
        step(i)
    dist.barrier()
    if use_gpu:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0

    # MAX over ranks
    t = torch.tensor([elapsed], dtype=torch.float64,
                     device=device if use_gpu else "cpu")
    if world_size > 1:
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
    elapsed = float(t.item())

    global_batch = args.micro_batch * args.gas * world_size
    tokens_per_step = global_batch * seq
    tokens_per_sec = tokens_per_step * args.steps / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    if rank == 0:
        result = {
            "metric": "tokens_per_sec",
            "value": round(tokens_per_sec, 1),
            "unit": "tokens/s",
            "n_gpus": world_size,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "n_params": n_params,
                "global_batch": global_batch,
                "seq_len": seq,
                "parallelism": f"zero{args.zero_stage}_dp{world_size}",
                "offload": bool(args.offload),
            },
        }
        print(json.dumps(result))


if __name__ == "__main__":
    main()
