import pytest
import torch

from deepspeed_amd.config import Config


def test_batch_reconciliation_full():
    c = Config({"train_batch_size": 16, "train_micro_batch_size_per_gpu": 2,
                "gradient_accumulation_steps": 4}, world_size=2)
    assert (c.train_batch_size, c.train_micro_batch_size_per_gpu,
            c.gradient_accumulation_steps) == (16, 2, 4)


def test_batch_reconciliation_infer_gas():
    c = Config({"train_batch_size": 16, "train_micro_batch_size_per_gpu": 2},
               world_size=2)
    assert c.gradient_accumulation_steps == 4


def test_batch_reconciliation_infer_tb():
    c = Config({"train_micro_batch_size_per_gpu": 3,
                "gradient_accumulation_steps": 5}, world_size=4)
    assert c.train_batch_size == 60


def test_batch_reconciliation_mismatch():
    with pytest.raises(ValueError):
        Config({"train_batch_size": 10, "train_micro_batch_size_per_gpu": 3,
                "gradient_accumulation_steps": 1}, world_size=2)


def test_dtype_selection():
    assert Config({"bf16": {"enabled": True}}).dtype == torch.bfloat16
    assert Config({"fp16": {"enabled": True}}).dtype == torch.float16
    assert Config({}).dtype == torch.float32
    with pytest.raises(ValueError):
        Config({"bf16": {"enabled": True}, "fp16": {"enabled": True}})


def test_zero_config_defaults():
    c = Config({"zero_optimization": {"stage": 2}})
    assert c.zero.stage == 2
    assert c.zero.reduce_bucket_size == 500_000_000
    assert c.zero.overlap_comm


def test_unknown_key_rejected():
    with pytest.raises(Exception):
        Config({"zero_optimization": {"stag": 2}})


def test_json_file_roundtrip(tmp_path):
    import json
    p = tmp_path / "ds.json"
    p.write_text(json.dumps({"train_batch_size": 8,
                             "optimizer": {"type": "AdamW",
                                           "params": {"lr": 0.1}}}))
    c = Config(str(p), world_size=1)
    assert c.optimizer.type == "AdamW"
    assert c.optimizer.params["lr"] == 0.1
