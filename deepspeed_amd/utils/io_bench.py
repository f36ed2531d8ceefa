"""Disk I/O micro-benchmark CLI (reference: bin/ds_io over csrc/aio):
measures the async engine's sequential write/read bandwidth.

    python -m deepspeed_amd.utils.io_bench --size-mb 512 --path /tmp
"""

import argparse
import os
import time

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--size-mb", type=int, default=256)
    ap.add_argument("--path", type=str, default="/tmp")
    ap.add_argument("--block-kb", type=int, default=1024)
    ap.add_argument("--threads", type=int, default=8)
    args = ap.parse_args()

    from ..ops._loader import get_ext
    ext = get_ext()
    if ext is None or not hasattr(ext, "AioHandle"):
        raise SystemExit("native aio op not built "
                         "(python setup.py build_ext --inplace)")
    h = ext.AioHandle(args.block_kb * 1024, args.threads)
    n = args.size_mb * 1024 * 1024
    t = torch.randint(0, 255, (n,), dtype=torch.uint8)
    p = os.path.join(args.path, "ds_io_bench.bin")

    t0 = time.perf_counter()
    h.async_pwrite(t, p)
    assert h.wait() == 0
    os.sync()
    tw = time.perf_counter() - t0

    r = torch.empty_like(t)
    t0 = time.perf_counter()
    h.async_pread(r, p)
    assert h.wait() == 0
    tr = time.perf_counter() - t0
    os.unlink(p)
    assert torch.equal(t[:1024], r[:1024])
    print(f"write: {n / tw / 1e9:.2f} GB/s   read: {n / tr / 1e9:.2f} GB/s "
          f"({args.size_mb} MiB, block {args.block_kb} KiB, "
          f"{args.threads} threads)")


if __name__ == "__main__":
    main()
