"""Hybrid engine — ZeRO-3 training + in-process generation for RLHF
(reference: deepspeed/runtime/hybrid_engine.py DeepSpeedHybridEngine :30,
_zero3_forward :362).

The reference gathers ZeRO-3 params layer-by-layer during generate because
A100s cannot hold a full replica. A 288 GB MI355X holds the whole bf16
model (even 70B = 140 GB), so this engine materializes ALL units once per
generate call — one big coalesced all-gather burst over xGMI instead of
per-layer latency-bound gathers every decode step — runs the KV-cached
loop, then drops back to sharded state.
"""

import contextlib

import torch

from .engine import Engine
from .zero.stage3 import ZeroStage3Optimizer


class DeepSpeedHybridEngine(Engine):
    """Training engine whose ``generate()`` serves the current weights."""

    def _materialized(self):
        if isinstance(self.optimizer, ZeroStage3Optimizer):
            params = [p for u in self.optimizer.units for p in u.params]
            return self.optimizer.gathered_params(params)
        return contextlib.nullcontext()

    @torch.no_grad()
    def generate(self, input_ids, max_new_tokens=32, do_sample=False,
                 temperature=1.0, top_k=0, eos_token_id=None):
        from ..inference.engine import kv_generate
        cfg = getattr(self.module, "cfg", None)
        assert cfg is not None, "generate() needs a model with .cfg geometry"
        was_training = self.module.training
        self.module.eval()
        try:
            with self._materialized():
                out = kv_generate(
                    self.module, input_ids.to(self.device),
                    n_layers=cfg.num_layers,
                    kv_heads=getattr(cfg, "num_kv_heads", None)
                    or cfg.num_heads,
                    head_dim=cfg.head_dim, max_seq=cfg.max_seq_len,
                    dtype=self.dtype if self.dtype != torch.float32
                    else torch.float32,
                    max_new_tokens=max_new_tokens, do_sample=do_sample,
                    temperature=temperature, top_k=top_k,
                    eos_token_id=eos_token_id)
        finally:
            if was_training:
                self.module.train()
        return out
