"""Collective micro-benchmark CLI (reference: bin/ds_bench ->
benchmarks/communication): measures all-reduce / all-gather /
reduce-scatter / all-to-all bus bandwidth over RCCL on the xGMI mesh.

Launch: python -m deepspeed_amd.launcher.runner --num_gpus 8 \
            -m --module deepspeed_amd.utils.comm_bench [--maxsize 28]
"""

import argparse
import os
import time

import torch

from .. import comm as dist


def _bw(nbytes, seconds, world, op):
    # bus bandwidth correction factors (NCCL convention)
    factor = {"all_reduce": 2 * (world - 1) / world,
              "all_gather": (world - 1) / world,
              "reduce_scatter": (world - 1) / world,
              "all_to_all": (world - 1) / world}[op]
    return nbytes * factor / seconds / 1e9


def run(op: str, numel: int, device, world, iters=20, warmup=5):
    t = torch.randn(numel, device=device, dtype=torch.bfloat16)
    out_full = torch.empty(numel * world, device=device, dtype=torch.bfloat16)
    out_shard = torch.empty(max(numel // world, 1), device=device,
                            dtype=torch.bfloat16)

    def call():
        if op == "all_reduce":
            dist.all_reduce(t)
        elif op == "all_gather":
            dist.all_gather_into_tensor(out_full, t)
        elif op == "reduce_scatter":
            dist.reduce_scatter_tensor(out_shard, t)
        elif op == "all_to_all":
            dist.all_to_all_single(torch.empty_like(t), t)

    for _ in range(warmup):
        call()
    if device.type == "cuda":
        torch.cuda.synchronize()
    dist.barrier()
    t0 = time.perf_counter()
    for _ in range(iters):
        call()
    if device.type == "cuda":
        torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    return dt, _bw(numel * 2, dt, world, op)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--minsize", type=int, default=20,
                    help="log2 of smallest message bytes")
    ap.add_argument("--maxsize", type=int, default=28)
    ap.add_argument("--ops", type=str,
                    default="all_reduce,all_gather,reduce_scatter,all_to_all")
    args = ap.parse_args()
    dist.init_distributed()
    world = dist.get_world_size()
    rank = dist.get_rank()
    device = torch.device(f"cuda:{int(os.environ.get('LOCAL_RANK', 0))}") \
        if torch.cuda.is_available() else torch.device("cpu")
    if device.type == "cuda":
        torch.cuda.set_device(device)
    for op in args.ops.split(","):
        if rank == 0:
            print(f"== {op} (world={world}) ==")
        for p in range(args.minsize, args.maxsize + 1, 2):
            numel = (1 << p) // 2  # bf16
            dt, bw = run(op, numel, device, world)
            if rank == 0:
                print(f"  {1 << p:>12d} B  {dt * 1e3:8.3f} ms  "
                      f"{bw:7.1f} GB/s busbw")


if __name__ == "__main__":
    main()
