"""Point-to-point activation/gradient exchange between adjacent stages
(reference: deepspeed/runtime/pipe/p2p.py :46-81, meta exchange
engine.py:_send_tensor_meta:928).

Tensors travel as-is over RCCL (GPU) or gloo (CPU tests). The first transfer
of each (peer, stream-slot) sends a small int64 header describing the tuple
structure (count, per-tensor ndim/shape/dtype/requires_grad); afterwards the
receiver reuses the cached meta and preallocates buffers, so steady-state
microbatches cost exactly one send per tensor.
"""

from typing import List, Sequence, Tuple

import torch

from ... import comm as dist

_DTYPES = [torch.float32, torch.float16, torch.bfloat16, torch.int64,
           torch.int32, torch.bool, torch.uint8, torch.float64]
_DTYPE_CODE = {d: i for i, d in enumerate(_DTYPES)}
_HEADER_LEN = 128


def _encode_meta(tensors: Sequence[torch.Tensor], device) -> torch.Tensor:
    h = torch.zeros(_HEADER_LEN, dtype=torch.int64)
    h[0] = len(tensors)
    i = 1
    for t in tensors:
        h[i] = t.dim()
        h[i + 1] = _DTYPE_CODE[t.dtype]
        h[i + 2] = int(t.requires_grad)
        for j, s in enumerate(t.shape):
            h[i + 3 + j] = s
        i += 3 + t.dim()
        assert i < _HEADER_LEN, "activation tuple too large for p2p header"
    return h.to(device)


def _decode_meta(h: torch.Tensor) -> List[Tuple]:
    h = h.cpu()
    n = int(h[0])
    metas, i = [], 1
    for _ in range(n):
        ndim = int(h[i])
        dtype = _DTYPES[int(h[i + 1])]
        requires_grad = bool(h[i + 2])
        shape = tuple(int(h[i + 3 + j]) for j in range(ndim))
        metas.append((shape, dtype, requires_grad))
        i += 3 + ndim
    return metas


class PipeP2P:
    """Per-engine p2p helper with cached metas keyed by (peer, slot).

    Sends are ISends: in 1F1B adjacent stages send to each other
    concurrently (my activation up, your grad down) and a synchronous send
    on both sides deadlocks. The handle AND the contiguous buffer are held
    in ``_pending`` until drained so the transport never reads freed memory.
    """

    MAX_PENDING = 32

    def __init__(self, device):
        self.device = device
        self._send_meta_done = set()
        self._recv_meta = {}
        self._pending = []  # (work_handle, tensor_kept_alive)

    def send(self, tensors, peer: int, slot: str):
        if torch.is_tensor(tensors):
            tensors = (tensors,)
        key = (peer, slot)
        if key not in self._send_meta_done:
            meta = _encode_meta(tensors, self.device)
            self._pending.append((dist.isend(meta, dst=peer), meta))
            self._send_meta_done.add(key)
        for t in tensors:
            buf = t.contiguous()
            self._pending.append((dist.isend(buf, dst=peer), buf))
        if len(self._pending) > self.MAX_PENDING:
            self.flush()

    def flush(self):
        for h, _buf in self._pending:
            h.wait()
        self._pending.clear()

    def recv(self, peer: int, slot: str):
        key = (peer, slot)
        if key not in self._recv_meta:
            h = torch.zeros(_HEADER_LEN, dtype=torch.int64, device=self.device)
            dist.recv(h, src=peer)
            self._recv_meta[key] = _decode_meta(h)
        out = []
        for shape, dtype, requires_grad in self._recv_meta[key]:
            buf = torch.empty(shape, dtype=dtype, device=self.device)
            dist.recv(buf, src=peer)
            buf.requires_grad_(requires_grad and dtype.is_floating_point)
            out.append(buf)
        return tuple(out)
