"""Autotuner (reference: deepspeed/autotuning/autotuner.py Autotuner :42,
tune :404).

Searches ZeRO stage x micro-batch-size for the best measured throughput.
The reference forks whole training jobs through the launcher; here the
experiment runner is injectable — the default builds an engine in-process
and times a few steps, which is what a single MI355X node needs; fleets can
pass a runner that shells out through the launcher instead.

Search strategy (reference "fast" mode): first find the largest micro batch
that does not OOM, then scan ZeRO stages at that batch, preferring lower
stages on ties (less comm).
"""

import json
import time
from typing import Callable, Dict, List, Optional

import torch

from ..utils.logging import logger


class Autotuner:
    def __init__(self,
                 base_config: Dict,
                 runner: Optional[Callable[[Dict], float]] = None,
                 micro_batch_sizes: Optional[List[int]] = None,
                 zero_stages: Optional[List[int]] = None,
                 model_factory: Optional[Callable] = None,
                 sample_factory: Optional[Callable[[int], tuple]] = None,
                 steps: int = 4):
        self.base_config = dict(base_config)
        self.micro_batch_sizes = micro_batch_sizes or [1, 2, 4, 8, 16]
        self.zero_stages = zero_stages if zero_stages is not None \
            else [0, 1, 2, 3]
        self.runner = runner or self._default_runner
        self.model_factory = model_factory
        self.sample_factory = sample_factory
        self.steps = steps
        self.results: List[Dict] = []

    # ------------------------------------------------------------- experiments
    def _default_runner(self, config: Dict) -> float:
        """Build an engine and time `steps` train steps; returns samples/s
        (0.0 on OOM)."""
        assert self.model_factory and self.sample_factory, \
            "default runner needs model_factory and sample_factory"
        import deepspeed_amd
        mb = config["train_micro_batch_size_per_gpu"]
        try:
            model = self.model_factory()
            engine, _, _, _ = deepspeed_amd.initialize(model=model,
                                                       config=dict(config))
            args, kwargs = self.sample_factory(mb)
            for _ in range(2):  # warmup
                loss = engine(*args, **kwargs)
                engine.backward(loss)
                engine.step()
            if torch.cuda.is_available():
                torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(self.steps):
                loss = engine(*args, **kwargs)
                engine.backward(loss)
                engine.step()
            if torch.cuda.is_available():
                torch.cuda.synchronize()
            dt = time.perf_counter() - t0
            return mb * self.steps / dt
        except torch.cuda.OutOfMemoryError:
            if torch.cuda.is_available():
                torch.cuda.empty_cache()
            return 0.0

    def _experiment(self, stage: int, mb: int) -> float:
        cfg = dict(self.base_config)
        cfg["train_micro_batch_size_per_gpu"] = mb
        zo = dict(cfg.get("zero_optimization", {}))
        zo["stage"] = stage
        cfg["zero_optimization"] = zo
        metric = self.runner(cfg)
        self.results.append({"stage": stage, "micro_batch": mb,
                             "metric": metric})
        logger.info(f"autotune: stage={stage} mb={mb} -> {metric:.2f}")
        return metric

    # ------------------------------------------------------------------- tune
    def tune(self) -> Dict:
        """Returns the best full config found."""
        stage0 = self.zero_stages[-1]  # most memory-lenient stage for probing
        best_mb, best_mb_metric = None, 0.0
        prev = 0.0
        for mb in self.micro_batch_sizes:
            m = self._experiment(stage0, mb)
            if m > best_mb_metric:
                best_mb, best_mb_metric = mb, m
            if m == 0.0 or (prev > 0 and m < prev * 1.02):
                break  # OOM or diminishing returns: stop growing
            prev = m
        assert best_mb is not None and best_mb_metric > 0, \
            "no experiment succeeded"

        best = {"stage": stage0, "mb": best_mb, "metric": best_mb_metric}
        for stage in self.zero_stages:
            if stage == stage0:
                continue
            m = self._experiment(stage, best_mb)
            if m > best["metric"] * 1.02:  # prefer existing on near-ties
                best = {"stage": stage, "mb": best_mb, "metric": m}

        cfg = dict(self.base_config)
        cfg["train_micro_batch_size_per_gpu"] = best["mb"]
        zo = dict(cfg.get("zero_optimization", {}))
        zo["stage"] = best["stage"]
        cfg["zero_optimization"] = zo
        self.best_metric = best["metric"]
        return cfg

    def write_results(self, path: str):
        with open(path, "w") as f:
            json.dump(self.results, f, indent=2)


# --------------------------------------------------------------- mem model

def estimate_memory_per_gpu(n_params: int, stage: int, world_size: int = 1,
                            micro_batch: int = 1, seq_len: int = 2048,
                            hidden: int = 4096, n_layers: int = 32,
                            dtype_bytes: int = 2, offload: bool = False,
                            activation_checkpointing: bool = False) -> int:
    """Model-based per-GPU memory estimate in bytes (reference
    autotuner model_info pruning, autotuner.py:~700): weights + grads +
    optimizer states partitioned per ZeRO stage, plus an activation term.
    Used to PRUNE experiments that cannot fit before running them."""
    P = n_params
    W = max(world_size, 1)
    weights = P * dtype_bytes / (W if stage >= 3 else 1)
    grads = P * dtype_bytes / (W if stage >= 2 else 1)
    # fp32 master + adam m/v = 12 bytes/param, partitioned from stage 1
    optim = P * 12 / (W if stage >= 1 else 1)
    if offload:
        optim = 0
    # activation memory: ~ mb * seq * hidden * layers * c (c~16 bytes dense,
    # ~2 with full activation checkpointing)
    act_c = 2 if activation_checkpointing else 16
    acts = micro_batch * seq_len * hidden * n_layers * act_c
    return int(weights + grads + optim + acts)


def prune_search_space(n_params: int, gpu_mem_bytes: int, stages, mbs,
                       world_size: int = 1, **model_kw):
    """Filter (stage, mb) combos whose estimated footprint exceeds the
    device memory (with a 10% headroom)."""
    keep = []
    budget = gpu_mem_bytes * 0.9
    for stage in stages:
        for mb in mbs:
            est = estimate_memory_per_gpu(n_params, stage, world_size, mb,
                                          **model_kw)
            if est <= budget:
                keep.append((stage, mb))
    return keep


# -------------------------------------------------------- forked experiments

def _subprocess_entry(q, base_config, stage, mb, model_factory,
                      sample_factory, steps):
    import os
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("LOCAL_RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", str(29600 + os.getpid() % 200))
    try:
        tuner = Autotuner(base_config, model_factory=model_factory,
                          sample_factory=sample_factory, steps=steps,
                          zero_stages=[stage], micro_batch_sizes=[mb])
        metric = tuner._experiment(stage, mb)
        q.put(("ok", metric))
    except BaseException as e:  # noqa: BLE001 — OOM/crash isolation
        q.put(("err", repr(e)))


class SubprocessRunner:
    """Run each experiment in a FORKED process (the reference launches
    whole jobs through the launcher, launcher/runner.py fork — same
    isolation property: an OOM or crash in one trial cannot poison the
    tuner process or the allocator)."""

    def __init__(self, model_factory, sample_factory, steps=4, timeout=300):
        self.model_factory = model_factory
        self.sample_factory = sample_factory
        self.steps = steps
        self.timeout = timeout

    def __call__(self, config: Dict) -> float:
        import multiprocessing as mp
        ctx = mp.get_context("fork" if not torch.cuda.is_available()
                             else "spawn")
        q = ctx.Queue()
        stage = config.get("zero_optimization", {}).get("stage", 0)
        mb = config["train_micro_batch_size_per_gpu"]
        p = ctx.Process(target=_subprocess_entry,
                        args=(q, config, stage, mb, self.model_factory,
                              self.sample_factory, self.steps))
        p.start()
        p.join(self.timeout)
        if p.is_alive():
            p.terminate()
            p.join(5)
            return 0.0
        if q.empty():
            return 0.0
        kind, val = q.get()
        return float(val) if kind == "ok" else 0.0


class GridAutotuner(Autotuner):
    """Exhaustive (stage x mb x gas) grid with memory-model pruning —
    the reference's "full"/"grid" tune mode on top of the greedy fast
    mode the base class implements."""

    def __init__(self, *args, gas_options=None, n_params=None,
                 gpu_mem_bytes=None, model_kw=None, **kw):
        super().__init__(*args, **kw)
        self.gas_options = gas_options or [1]
        self.n_params = n_params
        self.gpu_mem_bytes = gpu_mem_bytes
        self.model_kw = model_kw or {}

    def tune(self) -> Dict:
        combos = [(s, m) for s in self.zero_stages
                  for m in self.micro_batch_sizes]
        if self.n_params and self.gpu_mem_bytes:
            kept = prune_search_space(self.n_params, self.gpu_mem_bytes,
                                      self.zero_stages,
                                      self.micro_batch_sizes,
                                      **self.model_kw)
            pruned = len(combos) - len(kept)
            logger.info(f"autotune: pruned {pruned}/{len(combos)} combos by "
                        "the memory model")
            combos = kept
        best = None
        for stage, mb in combos:
            for gas in self.gas_options:
                cfg = dict(self.base_config)
                cfg["train_micro_batch_size_per_gpu"] = mb
                cfg["gradient_accumulation_steps"] = gas
                zo = dict(cfg.get("zero_optimization", {}))
                zo["stage"] = stage
                cfg["zero_optimization"] = zo
                metric = self.runner(cfg)
                self.results.append({"stage": stage, "micro_batch": mb,
                                     "gas": gas, "metric": metric})
                if metric > 0 and (best is None or metric > best[0]):
                    best = (metric, cfg)
        assert best is not None, "no experiment succeeded"
        self.best_metric = best[0]
        return best[1]
