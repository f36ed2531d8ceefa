"""Offline dataset analysis for curriculum learning (reference:
deepspeed/runtime/data_pipeline/data_analyzer.py DataAnalyzer :22).

Maps each sample to metric values (e.g. sequence length, vocab rarity),
writes an index file per metric, and builds the difficulty list the
curriculum DeepSpeedDataSampler consumes. Multi-worker sharding matches
the reference contract (worker_id/num_workers split, merged at the end).
"""

import json
import os
from typing import Callable, Dict, List, Sequence

import torch


def metric_seqlen(sample) -> float:
    """Default difficulty: token count (attention cost grows with it)."""
    if torch.is_tensor(sample):
        return float(sample.numel())
    if isinstance(sample, (tuple, list)):
        return float(len(sample[0]) if hasattr(sample[0], "__len__")
                     else len(sample))
    return float(len(sample))


def metric_vocab_rarity(vocab_freq: Dict[int, float]):
    """Curriculum metric: mean inverse token frequency."""
    def fn(sample):
        ids = sample.tolist() if torch.is_tensor(sample) else list(sample)
        if not ids:
            return 0.0
        return float(sum(1.0 / max(vocab_freq.get(int(t), 1.0), 1e-9)
                         for t in ids) / len(ids))
    return fn


class DataAnalyzer:
    def __init__(self, dataset: Sequence,
                 metric_names: List[str] = None,
                 metric_functions: List[Callable] = None,
                 save_path: str = "./data_analysis",
                 worker_id: int = 0, num_workers: int = 1):
        self.dataset = dataset
        self.metric_names = metric_names or ["seqlen"]
        self.metric_functions = metric_functions or [metric_seqlen]
        assert len(self.metric_names) == len(self.metric_functions)
        self.save_path = save_path
        self.worker_id = worker_id
        self.num_workers = num_workers

    def _shard_path(self, metric, worker):
        return os.path.join(self.save_path, f"{metric}_worker{worker}.json")

    def run_map(self):
        """Compute this worker's shard of every metric."""
        os.makedirs(self.save_path, exist_ok=True)
        n = len(self.dataset)
        results = {m: {} for m in self.metric_names}
        for i in range(self.worker_id, n, self.num_workers):
            s = self.dataset[i]
            for name, fn in zip(self.metric_names, self.metric_functions):
                results[name][i] = fn(s)
        for name in self.metric_names:
            with open(self._shard_path(name, self.worker_id), "w") as f:
                json.dump(results[name], f)
        return results

    def run_reduce(self):
        """Merge all workers' shards into index_to_metric files; returns
        {metric: difficulty list aligned with dataset order}."""
        merged = {}
        for name in self.metric_names:
            vals = {}
            for w in range(self.num_workers):
                with open(self._shard_path(name, w)) as f:
                    vals.update({int(k): v for k, v in json.load(f).items()})
            ordered = [vals[i] for i in range(len(self.dataset))]
            out = os.path.join(self.save_path, f"{name}_index_to_metric.json")
            with open(out, "w") as f:
                json.dump(ordered, f)
            # sample ids sorted by difficulty (reference metric_to_sample)
            by_metric = sorted(range(len(ordered)), key=lambda i: ordered[i])
            with open(os.path.join(self.save_path,
                                   f"{name}_metric_to_sample.json"), "w") as f:
                json.dump(by_metric, f)
            merged[name] = ordered
        return merged


def load_index_to_metric(save_path: str, metric: str) -> List[float]:
    with open(os.path.join(save_path, f"{metric}_index_to_metric.json")) as f:
        return json.load(f)


def _map_worker(args):
    (dataset, names, fns, save_path, wid, nw) = args
    DataAnalyzer(dataset, names, fns, save_path, wid, nw).run_map()


def run_analysis_parallel(dataset, metric_names=None, metric_functions=None,
                          save_path="./data_analysis", num_workers=4):
    """Map-reduce over `num_workers` PROCESSES (reference DataAnalyzer's
    multi-worker map, data_analyzer.py:22 run_map_reduce): each worker
    writes its metric shard, then one reduce merges + builds the
    index_to_metric / metric_to_sample files."""
    import multiprocessing as mp
    names = metric_names or ["seqlen"]
    fns = metric_functions or [metric_seqlen]
    ctx = mp.get_context("fork")
    with ctx.Pool(num_workers) as pool:
        pool.map(_map_worker,
                 [(dataset, names, fns, save_path, w, num_workers)
                  for w in range(num_workers)])
    return DataAnalyzer(dataset, names, fns, save_path, 0,
                        num_workers).run_reduce()
