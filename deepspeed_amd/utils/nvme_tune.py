"""NVMe configuration sweep (reference: bin/ds_nvme_tune + nvme/ perf
tools, ~1,283 LoC of sweep drivers): searches (block_size x thread_count)
for the aio engine, measures read and write bandwidth for each combo, and
prints the best settings as a ready-to-paste ``aio`` config block.

    python -m deepspeed_amd.utils.nvme_tune --path /local_nvme --size-mb 256
"""

import argparse
import json
import os
import time

import torch


def _bench_combo(ext, path, data, block_kb, threads):
    h = ext.AioHandle(block_kb * 1024, threads)
    p = os.path.join(path, f"nvme_tune_{block_kb}_{threads}.bin")
    try:
        t0 = time.perf_counter()
        h.async_pwrite(data, p)
        h.wait()
        tw = time.perf_counter() - t0
        out = torch.empty_like(data)
        t0 = time.perf_counter()
        h.async_pread(out, p)
        h.wait()
        tr = time.perf_counter() - t0
        n = data.numel()
        return n / tw / 1e9, n / tr / 1e9
    finally:
        if os.path.exists(p):
            os.unlink(p)


def tune(path=".", size_mb=256, block_kbs=(128, 256, 512, 1024, 4096),
         thread_counts=(1, 2, 4, 8, 16), verbose=True):
    from ..ops._loader import get_ext
    ext = get_ext()
    if ext is None or not hasattr(ext, "AioHandle"):
        raise SystemExit("native aio op not built")
    n = size_mb * 1024 * 1024
    data = torch.randint(0, 255, (n,), dtype=torch.uint8)
    results = []
    for bk in block_kbs:
        for th in thread_counts:
            w, r = _bench_combo(ext, path, data, bk, th)
            results.append({"block_kb": bk, "threads": th,
                            "write_GBps": round(w, 3),
                            "read_GBps": round(r, 3)})
            if verbose:
                print(f"block={bk:5d}K threads={th:2d}  "
                      f"write {w:6.2f} GB/s  read {r:6.2f} GB/s")
    best = max(results, key=lambda x: x["write_GBps"] + x["read_GBps"])
    cfg = {"aio": {"block_size": best["block_kb"] * 1024,
                   "thread_count": best["threads"],
                   "queue_depth": best["threads"] * 4,
                   "single_submit": False, "overlap_events": True}}
    if verbose:
        print("\nbest combo:", best)
        print("suggested config block:")
        print(json.dumps(cfg, indent=2))
    return results, cfg


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--path", type=str, default=".")
    ap.add_argument("--size-mb", type=int, default=256)
    ap.add_argument("--blocks", type=str, default="128,512,1024,4096")
    ap.add_argument("--threads", type=str, default="1,4,8,16")
    args = ap.parse_args()
    tune(args.path, args.size_mb,
         tuple(int(x) for x in args.blocks.split(",")),
         tuple(int(x) for x in args.threads.split(",")))


if __name__ == "__main__":
    main()
