"""Async I/O engine + tensor swapper tests (reference contract:
tests/unit/ops/aio/test_aio.py, runtime/swap_tensor tests)."""

import os

import pytest
import torch

from deepspeed_amd.ops._loader import get_ext

needs_ext = pytest.mark.skipif(
    get_ext() is None or not hasattr(get_ext(), "AioHandle"),
    reason="native aio op not built")


@needs_ext
@pytest.mark.parametrize("nbytes", [1024, (1 << 20) + 77, 8 << 20])
def test_aio_roundtrip(tmp_path, nbytes):
    ext = get_ext()
    h = ext.AioHandle(1 << 18, 4)
    t = torch.randint(0, 255, (nbytes,), dtype=torch.uint8)
    p = str(tmp_path / "blob.bin")
    h.async_pwrite(t, p)
    assert h.wait() == 0
    assert os.path.getsize(p) == nbytes
    r = torch.empty_like(t)
    h.async_pread(r, p)
    assert h.wait() == 0
    assert torch.equal(t, r)


@needs_ext
def test_aio_many_concurrent(tmp_path):
    ext = get_ext()
    h = ext.AioHandle(1 << 16, 8)
    tensors = [torch.randn(10000) for _ in range(16)]
    for i, t in enumerate(tensors):
        h.async_pwrite(t, str(tmp_path / f"t{i}.bin"))
    assert h.wait() == 0
    outs = [torch.empty(10000) for _ in range(16)]
    for i, o in enumerate(outs):
        h.async_pread(o, str(tmp_path / f"t{i}.bin"))
    assert h.wait() == 0
    for t, o in zip(tensors, outs):
        assert torch.equal(t, o)


@needs_ext
def test_tensor_swapper(tmp_path):
    from deepspeed_amd.runtime.swap_tensor import AsyncTensorSwapper
    sw = AsyncTensorSwapper(str(tmp_path))
    a = torch.randn(4, 1000)
    b = torch.randn(32, dtype=torch.float64)
    sw.swap_out("a", a)
    sw.swap_out("b", b)
    sw.synchronize()
    a2 = sw.swap_in("a")
    b2 = sw.swap_in("b")
    sw.synchronize()
    assert torch.equal(a, a2) and torch.equal(b, b2)
    assert a2.dtype == torch.float32 and b2.dtype == torch.float64
    sw.remove("a")
    assert not os.path.exists(str(tmp_path / "a.swp"))


def test_nvme_tune_sweep(tmp_path):
    """ds_nvme_tune equivalent: sweep produces per-combo bandwidths and a
    valid suggested aio config block."""
    from deepspeed_amd.utils.nvme_tune import tune
    results, cfg = tune(path=str(tmp_path), size_mb=4,
                        block_kbs=(128, 512), thread_counts=(1, 2),
                        verbose=False)
    assert len(results) == 4
    assert all(r["write_GBps"] > 0 and r["read_GBps"] > 0 for r in results)
    assert cfg["aio"]["block_size"] in (128 * 1024, 512 * 1024)
    assert cfg["aio"]["thread_count"] in (1, 2)


def test_aio_o_direct_roundtrip(tmp_path, monkeypatch):
    """DS_AIO_O_DIRECT=1: page-cache-bypassing mode (4K bounce buffers,
    rounded writes + exact-size truncate) round-trips arbitrary sizes;
    falls back transparently where the fs refuses O_DIRECT (tmpfs)."""
    import os
    import torch
    monkeypatch.setenv("DS_AIO_O_DIRECT", "1")
    from deepspeed_amd.ops._loader import get_ext
    ext = get_ext()
    if ext is None:
        return
    h = ext.AioHandle(64 * 1024, 4)
    for n in (4096 * 5, 4096 * 5 + 1337, 777):
        t = torch.randint(0, 255, (n,), dtype=torch.uint8)
        p = str(tmp_path / f"od_{n}.bin")
        h.async_pwrite(t, p)
        assert h.wait() == 0
        assert os.path.getsize(p) == n
        out = torch.empty_like(t)
        h.async_pread(out, p)
        assert h.wait() == 0
        assert torch.equal(out, t), n
