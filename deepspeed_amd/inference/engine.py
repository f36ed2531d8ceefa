"""Inference engine — TP sharding + KV-cache decoding + generate()
(reference: deepspeed/inference/engine.py InferenceEngine :40,
init_inference deepspeed/__init__.py:291).

MI355X-first shape: the model's hot ops are already the framework's HIP
kernels (RMSNorm/RoPE/SwiGLU) and hipBLASLt GEMMs; this engine adds the
serving mechanics — tensor-parallel sharding over the xGMI mesh
(see auto_tp.py), a contiguous KV cache, and a sampling loop.
"""

import os
from dataclasses import dataclass, field
from typing import Optional

import torch
import torch.nn as nn

from .. import accel
from .. import comm as dist
from ..parallel import groups
from ..utils.logging import log_dist
from .auto_tp import shard_attention_heads, shard_model
from .kv_cache import StaticKVCache


@dataclass
class InferenceConfig:
    dtype: object = torch.bfloat16
    tensor_parallel: dict = field(default_factory=dict)  # {"tp_size": N}
    max_out_tokens: int = 1024
    min_out_tokens: int = 1
    replace_with_kernel_inject: bool = False  # model already uses HIP ops
    checkpoint: Optional[str] = None
    enable_cuda_graph: bool = False  # hipGraph capture of the decode step

    @property
    def tp_size(self) -> int:
        if isinstance(self.tensor_parallel, dict):
            return int(self.tensor_parallel.get("tp_size", 1))
        return int(getattr(self.tensor_parallel, "tp_size", 1))


class InferenceEngine(nn.Module):
    def __init__(self, model: nn.Module, config: InferenceConfig):
        super().__init__()
        self.module = model
        self.config = config
        self._kv = None
        self._graph = None

        if config.replace_with_kernel_inject:
            from ..module_inject import (replace_transformer_layer,
                                         HFInjectionPolicy)
            self.injection_policy = HFInjectionPolicy()
            replace_transformer_layer(self.module, self.injection_policy)

        if config.checkpoint:
            self._load_checkpoint(config.checkpoint)

        tp = config.tp_size
        if tp > 1:
            if not dist.is_initialized():
                dist.init_distributed()
            groups.initialize_tensor_parallel(tp)
            self.tp_group = groups.get_tensor_parallel_group()
            self.tp_rank = groups.get_tensor_parallel_rank()
            n = shard_model(self.module, self.tp_group, self.tp_rank, tp)
            shard_attention_heads(self.module, self.tp_rank, tp)
            log_dist(f"AutoTP: sharded {n} linears over tp={tp}")
        else:
            self.tp_group = None
            self.tp_rank = 0

        self._weight_quantized = False
        # NOTE: weight-only quantization targets plain nn.Linear modules;
        # with tp_size > 1 the TP-sharded linear wrappers keep their 16-bit
        # shards (quantized TP serving = follow-up, docs/roadmap.md)
        if isinstance(config.dtype, str) and config.dtype in ("fp4", "fp6",
                                                              "fp12"):
            # FP6-style weight-only float quantization (reference
            # DeepSpeed-FP6 / fp_quantizer): tight-packed weights + group
            # scales on GPU via fp_quant.hip; bit-accurate emulation on CPU
            bits = int(config.dtype[2:])
            self.module.to(torch.bfloat16)
            if accel.available():
                # quantize ON device so the packed kernel (not the dense
                # emulation) produces the resident weights
                self.module.to(accel.current_device())
            config.dtype = torch.bfloat16
            n = self._quantize_linear_weights_fp(bits)
            self._weight_quantized = True
            log_dist(f"init_inference: fp{bits} weight-only quantized "
                     f"{n} linears")
        if isinstance(config.dtype, str):
            config.dtype = {"fp32": torch.float32, "fp16": torch.float16,
                            "bf16": torch.bfloat16,
                            "int8": torch.int8}[config.dtype]
        if config.dtype == torch.int8:
            # weight-only int8 (reference init_inference(dtype=torch.int8)
            # -> GroupQuantizer): int8 weights + group scales resident,
            # bf16 activations, dequant on the fly per linear
            self.module.to(torch.bfloat16)
            if accel.available():
                self.module.to(accel.current_device())
            config.dtype = torch.bfloat16
            n = self._quantize_linear_weights()
            self._weight_quantized = True
            log_dist(f"init_inference: int8 weight-only quantized "
                     f"{n} linears")
        elif config.dtype != torch.float32:
            self.module.to(config.dtype)
        if accel.available():
            self.module.to(accel.current_device())
        self.device = next(self.module.parameters()).device
        self.module.eval()

    def _quantize_linear_weights(self, group_size: int = 2048) -> int:
        from ..linear.optimized_linear import (QuantizationConfig,
                                               QuantizedParameter)
        qcfg = QuantizationConfig(group_size=group_size)

        class WOQLinear(nn.Module):
            def __init__(self, lin: nn.Linear):
                super().__init__()
                self.qweight = QuantizedParameter(lin.weight.data, qcfg)
                self.bias = lin.bias

            def forward(self, x):
                w = self.qweight.dequantized().to(x.dtype)
                return torch.nn.functional.linear(x, w, self.bias)

        n = 0
        for parent in list(self.module.modules()):
            for name, child in list(parent.named_children()):
                if type(child) is nn.Linear:
                    setattr(parent, name, WOQLinear(child))
                    n += 1
        return n

    def _quantize_linear_weights_fp(self, bits: int,
                                    group_size: int = 2048) -> int:
        from ..ops import fp_quantizer as fpq
        from ..ops._loader import get_ext

        class FPWOQLinear(nn.Module):
            def __init__(self, lin: nn.Linear):
                super().__init__()
                self.bits = bits
                self.group_size = group_size
                self.shape = lin.weight.shape
                w = lin.weight.data
                if w.is_cuda and get_ext() is not None:
                    q, s = fpq.fp_quantize(w, bits, group_size)
                    self.register_buffer("q", q)
                    self.register_buffer("scales", s)
                    self.weight_emu = None
                else:
                    # CPU: identical numerics, dense storage (the packed
                    # layout is the GPU kernel's memory-format win)
                    self.weight_emu = nn.Parameter(
                        fpq.fp_emulate_reference(w.float(), bits,
                                                 group_size).to(w.dtype),
                        requires_grad=False)
                self.bias = lin.bias

            def forward(self, x):
                if self.weight_emu is not None:
                    w = self.weight_emu.to(x.dtype)
                else:
                    w = fpq.fp_dequantize(
                        self.q, self.scales,
                        int(self.shape[0] * self.shape[1]), self.bits,
                        self.group_size, x.dtype).view(self.shape)
                return torch.nn.functional.linear(x, w, self.bias)

        n = 0
        for parent in list(self.module.modules()):
            for name, child in list(parent.named_children()):
                if type(child) is nn.Linear:
                    setattr(parent, name, FPWOQLinear(child))
                    n += 1
        return n

    def _load_checkpoint(self, ckpt):
        """Load weights named by ``config.checkpoint`` (reference
        inference/engine.py _load_checkpoint): a state-dict file
        (.pt/.bin/.safetensors), a directory of such shards, or a JSON
        manifest {"checkpoints": [paths...]}. A model constructed on the
        meta device is materialized empty first, so rank memory never
        holds more than one shard + the final weights."""
        import glob as _glob
        import json

        paths = []
        if isinstance(ckpt, (list, tuple)):
            paths = list(ckpt)
        elif os.path.isdir(ckpt):
            for pat in ("*.safetensors", "*.pt", "*.bin"):
                paths += sorted(_glob.glob(os.path.join(ckpt, pat)))
        elif ckpt.endswith(".json"):
            with open(ckpt) as f:
                manifest = json.load(f)
            base = os.path.dirname(ckpt)
            paths = [p if os.path.isabs(p) else os.path.join(base, p)
                     for p in manifest["checkpoints"]]
        else:
            paths = [ckpt]
        if not paths:
            raise FileNotFoundError(f"no checkpoint files under {ckpt}")

        if any(p.is_meta for p in self.module.parameters()):
            self.module.to_empty(device="cpu")

        missing = set(n for n, _ in self.module.named_parameters())
        for path in paths:
            if path.endswith(".safetensors"):
                from safetensors.torch import load_file
                sd = load_file(path)
            else:
                sd = torch.load(path, map_location="cpu", weights_only=True)
                if isinstance(sd, dict) and "module" in sd and \
                        isinstance(sd["module"], dict):
                    sd = sd["module"]  # engine.save_checkpoint layout
            self.module.load_state_dict(sd, strict=False)
            missing -= set(sd)
        if missing:
            log_dist(f"init_inference checkpoint: {len(missing)} params not "
                     f"found in {len(paths)} file(s), e.g. "
                     f"{sorted(missing)[:3]}")

    def forward(self, *args, **kwargs):
        with torch.no_grad():
            return self.module(*args, **kwargs)

    # --------------------------------------------------------------- generate

    def _model_geometry(self):
        m = self.module
        cfg = getattr(m, "cfg", None)
        assert cfg is not None, "generate() needs a model with .cfg geometry"
        kv_heads = getattr(cfg, "num_kv_heads", None) or cfg.num_heads
        tp = self.config.tp_size
        return (cfg.num_layers, kv_heads // max(tp, 1) if tp > 1 else kv_heads,
                cfg.head_dim, cfg.max_seq_len)

    @torch.no_grad()
    def generate(self, input_ids: torch.Tensor, max_new_tokens: int = 32,
                 do_sample: bool = False, temperature: float = 1.0,
                 top_k: int = 0, top_p: float = 1.0,
                 eos_token_id: Optional[int] = None):
        """KV-cached autoregressive generation (greedy or sampled).

        input_ids: [B, S] prompt. Returns [B, S + new] including the prompt.
        """
        layers, kv_heads, head_dim, max_seq = self._model_geometry()
        if self.config.enable_cuda_graph and self.device.type == "cuda":
            from .graph import graph_generate
            return graph_generate(
                self.module, input_ids.to(self.device), n_layers=layers,
                kv_heads=kv_heads, head_dim=head_dim, max_seq=max_seq,
                dtype=(self.config.dtype if self.config.dtype
                       != torch.float32 else torch.float32),
                max_new_tokens=max_new_tokens, do_sample=do_sample,
                temperature=temperature, top_k=top_k, top_p=top_p,
                eos_token_id=eos_token_id)
        return kv_generate(self.module, input_ids.to(self.device),
                           n_layers=layers, kv_heads=kv_heads,
                           head_dim=head_dim, max_seq=max_seq,
                           dtype=(self.config.dtype
                                  if self.config.dtype != torch.float32
                                  else torch.float32),
                           max_new_tokens=max_new_tokens,
                           do_sample=do_sample, temperature=temperature,
                           top_k=top_k, top_p=top_p,
                           eos_token_id=eos_token_id)


def _select_token(logits, do_sample, temperature, top_k, top_p=1.0):
    if not do_sample:
        return logits.argmax(dim=-1, keepdim=True)
    logits = logits.float() / max(temperature, 1e-5)
    if top_k > 0:
        kth = logits.topk(top_k, dim=-1).values[..., -1, None]
        logits = logits.masked_fill(logits < kth, float("-inf"))
    if top_p < 1.0:
        # nucleus sampling: keep the smallest prefix of the sorted
        # distribution whose mass reaches top_p (always >= 1 token)
        sorted_logits, sorted_idx = logits.sort(dim=-1, descending=True)
        sorted_probs = torch.softmax(sorted_logits, dim=-1)
        cum = sorted_probs.cumsum(dim=-1)
        # drop token i if the mass BEFORE it already covers top_p
        drop_sorted = (cum - sorted_probs) >= top_p
        drop_sorted[..., 0] = False
        drop = torch.zeros_like(drop_sorted).scatter(-1, sorted_idx,
                                                     drop_sorted)
        logits = logits.masked_fill(drop, float("-inf"))
    probs = torch.softmax(logits, dim=-1)
    return torch.multinomial(probs, 1)


@torch.no_grad()
def kv_generate(module, input_ids, *, n_layers, kv_heads, head_dim, max_seq,
                dtype, max_new_tokens=32, do_sample=False, temperature=1.0,
                top_k=0, top_p=1.0, eos_token_id=None):
    """Shared KV-cached generation loop (used by InferenceEngine and the
    hybrid RLHF engine)."""
    device = input_ids.device
    B, S = input_ids.shape
    total = min(S + max_new_tokens, max_seq)
    kv = StaticKVCache(n_layers, B, kv_heads, total, head_dim, dtype=dtype,
                       device=device)
    out = input_ids
    positions = torch.arange(S, device=device,
                             dtype=torch.int32).expand(B, S).contiguous()
    logits = module(input_ids, positions=positions, kv_cache=kv)
    kv.advance()
    next_tok = _select_token(logits[:, -1], do_sample, temperature, top_k,
                             top_p)
    out = torch.cat([out, next_tok], dim=1)
    finished = torch.zeros(B, dtype=torch.bool, device=device)

    for _ in range(max_new_tokens - 1):
        if out.size(1) >= total:
            break
        if eos_token_id is not None:
            finished |= next_tok.squeeze(1) == eos_token_id
            if bool(finished.all()):
                break
        pos = torch.full((B, 1), kv.cur_len, device=device, dtype=torch.int32)
        logits = module(next_tok, positions=pos, kv_cache=kv)
        kv.advance()
        next_tok = _select_token(logits[:, -1], do_sample, temperature,
                                 top_k, top_p)
        out = torch.cat([out, next_tok], dim=1)
    return out
