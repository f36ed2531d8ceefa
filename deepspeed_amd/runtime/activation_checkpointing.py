"""Activation checkpointing (recompute) for MI355X.

Capability parity with the reference's
``deepspeed/runtime/activation_checkpointing/checkpointing.py``
(CheckpointFunction :488, RNG tracker :124, CPU-offloaded activations
:377-486) — re-designed for ROCm:

* One custom autograd ``Function`` saves the layer inputs, drops all
  intermediate activations, and recomputes the layer in backward with the
  forward's RNG state restored (CPU + device generators), so dropout-style
  ops replay identically.
* ``cpu_checkpointing`` ships the *saved inputs* to pinned host memory with
  ``copy_(non_blocking=True)`` on a dedicated side HIP stream, overlapping
  the D2H with the rest of forward; backward prefetches them H2D on the
  same side stream before the recompute needs them. On MI355X this trades
  ≈8 TB/s HBM3E for PCIe only for the per-layer boundary tensor (B·S·H),
  which is tiny next to the freed intermediate activations.
* ``partition_activations`` shards the saved boundary tensor across the
  model-parallel group (flat shard per rank, all-gather on recompute) —
  with 288 GB HBM per GPU this matters only for very long sequences, but
  the reference exposes it so we do too.
"""

from typing import Optional

import torch

from .. import comm as dist

_config = None
_side_stream: Optional[torch.cuda.Stream] = None
_mp_group = None


def configure(activation_config=None, mpu=None):
    global _config, _mp_group
    _config = activation_config
    if mpu is not None:
        _mp_group = mpu.get_model_parallel_group()


def _cfg(name, default=False):
    return getattr(_config, name, default) if _config is not None else default


def _get_side_stream():
    global _side_stream
    if _side_stream is None and torch.cuda.is_available():
        _side_stream = torch.cuda.Stream()
    return _side_stream


class _RNGState:
    """Snapshot/restore of CPU + current-device RNG (reference RNG tracker
    ``get_cuda_rng_tracker`` — flattened: we need fork/join, not named
    states, because TP weight init uses its own path in this framework)."""

    def __init__(self, device):
        self.cpu_state = torch.get_rng_state()
        self.device = device
        self.device_state = (torch.cuda.get_rng_state(device)
                             if device.type == "cuda" else None)

    def restore(self):
        torch.set_rng_state(self.cpu_state)
        if self.device_state is not None:
            torch.cuda.set_rng_state(self.device_state, self.device)


def _shard(t: torch.Tensor, group):
    ws = dist.get_world_size(group)
    rank = dist.get_rank(group)
    flat = t.reshape(-1)
    pad = (ws - flat.numel() % ws) % ws
    if pad:
        flat = torch.nn.functional.pad(flat, (0, pad))
    per = flat.numel() // ws
    return flat[rank * per:(rank + 1) * per].clone(), pad


def _unshard(shard: torch.Tensor, pad: int, shape, group):
    ws = dist.get_world_size(group)
    full = torch.empty(ws * shard.numel(), dtype=shard.dtype,
                       device=shard.device)
    dist.all_gather_into_tensor(full, shard.contiguous(), group=group)
    if pad:
        full = full[:-pad]
    return full.view(shape)


class CheckpointFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, run_function, *args):
        ctx.run_function = run_function
        device = None
        for a in args:
            if torch.is_tensor(a) and a.is_cuda:
                device = a.device
                break
        ctx.rng = _RNGState(device or torch.device("cpu"))

        with torch.no_grad():
            outputs = run_function(*args)

        cpu_ckpt = _cfg("cpu_checkpointing") and device is not None
        part_act = (_cfg("partition_activations") and _mp_group is not None
                    and dist.get_world_size(_mp_group) > 1)
        ctx.cpu_ckpt = cpu_ckpt
        ctx.part_act = part_act
        saved, meta = [], []
        stream = _get_side_stream() if cpu_ckpt else None
        if stream is not None:
            stream.wait_stream(torch.cuda.current_stream())
        for a in args:
            if not torch.is_tensor(a):
                meta.append(("obj", a))
                continue
            if part_act and a.is_floating_point() and a.requires_grad:
                sh, pad = _shard(a.detach(), _mp_group)
                meta.append(("shard", (a.shape, pad, a.requires_grad, a.dtype)))
                saved.append(sh)
            elif cpu_ckpt and a.is_cuda and a.is_floating_point():
                host = torch.empty_like(a, device="cpu", pin_memory=True)
                with torch.cuda.stream(stream):
                    host.copy_(a.detach(), non_blocking=True)
                meta.append(("cpu", (a.device, a.requires_grad)))
                saved.append(host)
            else:
                meta.append(("tensor", a.requires_grad))
                saved.append(a.detach())
        if stream is not None:
            # forward may free `a` after return; block the free until D2H done
            torch.cuda.current_stream().wait_stream(stream)
        ctx.meta = meta
        ctx.save_for_backward(*saved)
        return outputs

    @staticmethod
    def backward(ctx, *grad_outputs):
        saved = list(ctx.saved_tensors)
        stream = _get_side_stream() if ctx.cpu_ckpt else None
        args = []
        it = iter(saved)
        for kind, info in ctx.meta:
            if kind == "obj":
                args.append(info)
                continue
            t = next(it)
            if kind == "shard":
                shape, pad, req, dtype = info
                t = _unshard(t, pad, shape, _mp_group)
                t.requires_grad_(req)
            elif kind == "cpu":
                device, req = info
                dev_t = torch.empty_like(t, device=device)
                with torch.cuda.stream(stream):
                    dev_t.copy_(t, non_blocking=True)
                torch.cuda.current_stream().wait_stream(stream)
                t = dev_t.requires_grad_(req)
            else:
                t = t.detach().requires_grad_(info)
            args.append(t)

        outer = _RNGState(ctx.rng.device)
        ctx.rng.restore()
        with torch.enable_grad():
            outputs = ctx.run_function(*args)
        outer.restore()

        if torch.is_tensor(outputs):
            outputs = (outputs,)
        out_and_grad = [(o, g) for o, g in zip(outputs, grad_outputs)
                        if torch.is_tensor(o) and o.requires_grad]
        torch.autograd.backward([o for o, _ in out_and_grad],
                                [g for _, g in out_and_grad])
        grads = tuple(a.grad if torch.is_tensor(a) and a.requires_grad else None
                      for a in args)
        return (None,) + grads


def checkpoint(function, *args):
    """Recompute `function(*args)` in backward instead of storing
    intermediates (drop-in for the reference's
    ``deepspeed.checkpointing.checkpoint``)."""
    return CheckpointFunction.apply(function, *args)


def is_configured():
    return _config is not None
