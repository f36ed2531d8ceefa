"""Elastic training config math (reference: deepspeed/elasticity/
elasticity.py compute_elastic_config :233, _get_compatible_gpus).

Given micro-batch candidates and a max acceptable batch size, compute the
set of total batch sizes + GPU counts that keep batch size constant across
allowed scale points, so a job can resume at a different world size without
changing optimization behavior.
"""

from typing import Dict, List

LATEST_ELASTICITY_VERSION = 0.2


class ElasticityError(Exception):
    pass


def get_valid_gpus(batch_size: int, micro_batches: List[int],
                   min_valid_gpus: int, max_valid_gpus: int) -> List[int]:
    valid = set()
    for mb in micro_batches:
        if batch_size % mb:
            continue
        max_gpus = batch_size // mb
        for g in range(1, max_gpus + 1):
            if max_gpus % g == 0 and min_valid_gpus <= g <= max_valid_gpus:
                valid.add(g)
    return sorted(valid)


def get_best_candidates(candidate_batch_sizes: List[int],
                        micro_batches: List[int], min_gpus: int,
                        max_gpus: int, prefer_larger: bool):
    max_valid = 0
    best_bs = -1
    best_gpus = []
    for bs in candidate_batch_sizes:
        gpus = get_valid_gpus(bs, micro_batches, min_gpus, max_gpus)
        if len(gpus) > max_valid or (len(gpus) == max_valid and
                                     ((prefer_larger and bs > best_bs) or
                                      (not prefer_larger and bs < best_bs))):
            max_valid = len(gpus)
            best_bs = bs
            best_gpus = gpus
    return best_bs, best_gpus


def _candidate_batch_sizes(base_list: List[int], max_acceptable: int):
    candidates = set()
    for base in base_list:
        if base > max_acceptable:
            continue
        value = base
        while value <= max_acceptable:
            candidates.add(value)
            value *= 2
    return sorted(candidates)


def compute_elastic_config(ds_config: Dict, target_deepspeed_version: str = "",
                           world_size: int = 0, return_microbatch: bool = False):
    """Returns (final_batch_size, valid_gpus[, micro_batch])
    (reference elasticity.py:233)."""
    elastic = ds_config.get("elasticity", {})
    if not elastic.get("enabled", False):
        raise ElasticityError("elasticity not enabled in config")
    max_acceptable = int(elastic.get("max_train_batch_size", 2000))
    micro_batches = [int(m) for m in elastic.get("micro_batch_sizes",
                                                 [2, 4, 6])]
    min_gpus = int(elastic.get("min_gpus", 1))
    max_gpus = int(elastic.get("max_gpus", 10000))
    prefer_larger = bool(elastic.get("prefer_larger_batch", True))
    if not micro_batches or min(micro_batches) <= 0:
        raise ElasticityError(f"invalid micro_batch_sizes {micro_batches}")

    candidates = _candidate_batch_sizes(micro_batches, max_acceptable)
    final_batch, valid_gpus = get_best_candidates(
        candidates, micro_batches, min_gpus, max_gpus, prefer_larger)
    if final_batch <= 0:
        raise ElasticityError("no valid batch size found")

    if world_size > 0 and world_size not in valid_gpus:
        raise ElasticityError(
            f"world size {world_size} not in valid GPU counts {valid_gpus}")

    if return_microbatch or world_size > 0:
        mbs = [m for m in sorted(micro_batches, reverse=prefer_larger)
               if world_size > 0 and final_batch % (m * world_size) == 0]
        micro = mbs[0] if mbs else None
        if world_size > 0 and micro is None:
            raise ElasticityError(
                f"no micro batch fits batch {final_batch} at ws {world_size}")
        if return_microbatch:
            return final_batch, valid_gpus, micro
    return final_batch, valid_gpus
