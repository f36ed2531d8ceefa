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
