"""Autotuner tests (reference contract: tests/unit/autotuning/test_autotuning.py)."""

import torch


def _tiny_model():
    from deepspeed_amd.models import GPT2ForCausalLM, gpt2_tiny
    torch.manual_seed(0)
    return GPT2ForCausalLM(gpt2_tiny())


def _sample(mb):
    torch.manual_seed(1)
    ids = torch.randint(0, 128, (mb, 16))
    return (ids,), {"labels": ids}


def test_autotuner_picks_best_with_fake_runner():
    from deepspeed_amd.autotuning import Autotuner

    # synthetic cost model: throughput grows with mb until OOM at 16;
    # stage 2 is best at the chosen mb
    def runner(cfg):
        mb = cfg["train_micro_batch_size_per_gpu"]
        stage = cfg["zero_optimization"]["stage"]
        if mb >= 16:
            return 0.0  # OOM
        base = {0: 90.0, 1: 95.0, 2: 120.0, 3: 100.0}[stage]
        return base * min(mb, 8) / 8

    tuner = Autotuner({"optimizer": {"type": "AdamW", "params": {}}},
                      runner=runner)
    best = tuner.tune()
    assert best["train_micro_batch_size_per_gpu"] == 8
    assert best["zero_optimization"]["stage"] == 2
    assert tuner.best_metric == 120.0
    assert len(tuner.results) >= 5


def test_autotuner_end_to_end_inprocess():
    """Default runner: real engine steps on a tiny model (CPU)."""
    from .common import run_local

    def worker(rank, world):
        from deepspeed_amd.autotuning import Autotuner
        from deepspeed_amd.models import GPT2ForCausalLM, gpt2_tiny

        def model_factory():
            torch.manual_seed(0)
            return GPT2ForCausalLM(gpt2_tiny())

        def sample_factory(mb):
            ids = torch.randint(0, 128, (mb, 16))
            return (ids,), {"labels": ids}

        tuner = Autotuner(
            {"optimizer": {"type": "AdamW", "params": {"lr": 1e-4}}},
            micro_batch_sizes=[1, 2], zero_stages=[1, 2],
            model_factory=model_factory, sample_factory=sample_factory,
            steps=2)
        best = tuner.tune()
        assert best["train_micro_batch_size_per_gpu"] in (1, 2)
        assert best["zero_optimization"]["stage"] in (1, 2)
        assert tuner.best_metric > 0

    run_local(worker)


def test_memory_model_pruning():
    from deepspeed_amd.autotuning.autotuner import (estimate_memory_per_gpu,
                                                    prune_search_space)
    P = 8_000_000_000
    # 8B params, stage 3 over 8 GPUs fits a 288 GB device; stage 0 does not
    # fit mb=8 on a 24 GB device
    s3 = estimate_memory_per_gpu(P, 3, world_size=8, micro_batch=8,
                                 seq_len=4096, hidden=4096, n_layers=32)
    assert s3 < 288e9
    s0 = estimate_memory_per_gpu(P, 0, world_size=8, micro_batch=8,
                                 seq_len=4096, hidden=4096, n_layers=32)
    assert s0 > 24e9
    kept = prune_search_space(P, int(288e9), [0, 1, 2, 3], [1, 8],
                              world_size=8, seq_len=4096, hidden=4096,
                              n_layers=32)
    assert (3, 8) in kept
    # offload shrinks the optimizer term
    off = estimate_memory_per_gpu(P, 2, world_size=1, offload=True)
    non = estimate_memory_per_gpu(P, 2, world_size=1, offload=False)
    assert off < non


def test_grid_autotuner_with_subprocess_runner():
    from deepspeed_amd.autotuning.autotuner import (GridAutotuner,
                                                    SubprocessRunner)
    runner = SubprocessRunner(_tiny_model, _sample, steps=2, timeout=120)
    tuner = GridAutotuner({"train_micro_batch_size_per_gpu": 2,
                           "optimizer": {"type": "AdamW",
                                         "params": {"lr": 1e-3}}},
                          runner=runner, zero_stages=[0, 1],
                          micro_batch_sizes=[2], gas_options=[1, 2])
    best = tuner.tune()
    assert best["train_micro_batch_size_per_gpu"] == 2
    assert len(tuner.results) == 4
    assert tuner.best_metric > 0
