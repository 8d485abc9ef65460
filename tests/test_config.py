import pytest

from deepspeed_amd.config import DeepSpeedConfig


def test_batch_triple_full():
    c = DeepSpeedConfig({"train_batch_size": 16,
                         "train_micro_batch_size_per_gpu": 2,
                         "gradient_accumulation_steps": 4}, world_size=2)
    assert c.train_batch_size == 16


def test_batch_triple_invalid():
    with pytest.raises(ValueError):
        DeepSpeedConfig({"train_batch_size": 16,
                         "train_micro_batch_size_per_gpu": 3,
                         "gradient_accumulation_steps": 4}, world_size=2)


def test_batch_derivation():
    c = DeepSpeedConfig({"train_micro_batch_size_per_gpu": 2,
                         "gradient_accumulation_steps": 4}, world_size=2)
    assert c.train_batch_size == 16
    c = DeepSpeedConfig({"train_batch_size": 8}, world_size=2)
    assert c.train_micro_batch_size_per_gpu == 4
    assert c.gradient_accumulation_steps == 1


def test_zero_defaults():
    c = DeepSpeedConfig({"zero_optimization": {"stage": 3}})
    z = c.zero_config
    assert z.stage == 3
    assert z.reduce_bucket_size == int(5e8)
    assert z.prefetch_bucket_size == int(2e8)
    assert z.param_persistence_threshold == int(1e5)
    assert z.overlap_comm is True


def test_zero_aliases():
    c = DeepSpeedConfig({"zero_optimization": {
        "stage": 3, "stage3_prefetch_bucket_size": 123,
        "stage3_param_persistence_threshold": 456}})
    assert c.zero_config.prefetch_bucket_size == 123
    assert c.zero_config.param_persistence_threshold == 456


def test_dtype():
    import torch
    assert DeepSpeedConfig({"bf16": {"enabled": True}}).dtype == torch.bfloat16
    assert DeepSpeedConfig({"fp16": {"enabled": True}}).dtype == torch.float16
    assert DeepSpeedConfig({}).dtype == torch.float32
    with pytest.raises(ValueError):
        DeepSpeedConfig({"fp16": {"enabled": True}, "bf16": {"enabled": True}})
