"""Mixtral (MoE) model: single-proc training + 2-rank EP with ZeRO-1."""
import torch

from tests.common import run_distributed


def test_mixtral_tiny_cpu_trains():
    from deepspeed_amd.models.mixtral import MIXTRAL_CONFIGS, MixtralForCausalLM
    cfg = MIXTRAL_CONFIGS["mixtral-tiny"]
    torch.manual_seed(0)
    model = MixtralForCausalLM(cfg).float()
    opt = torch.optim.AdamW(model.parameters(), lr=1e-3)
    data = torch.randint(0, cfg.vocab_size, (2, 32))
    losses = []
    for _ in range(5):
        opt.zero_grad()
        loss = model(data, labels=data)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0]


def _mixtral_ep2(steps=4):
    import torch.distributed as tdist
    import deepspeed_amd
    from deepspeed_amd.comm import groups
    from deepspeed_amd.models.mixtral import (MIXTRAL_CONFIGS,
                                              MixtralForCausalLM)
    groups.reset_groups()
    rank = tdist.get_rank()
    cfg = MIXTRAL_CONFIGS["mixtral-tiny"]
    cfg.ep_size = 2
    torch.manual_seed(100 + rank)
    model = MixtralForCausalLM(cfg)
    config = {
        "train_micro_batch_size_per_gpu": 2,
        "optimizer": {"type": "AdamW", "params": {"lr": 3e-4}},
        "zero_optimization": {"stage": 1, "reduce_bucket_size": 100000},
        "bf16": {"enabled": True},
    }
    # EP groups must exist before the optimizer builds expert buckets
    groups.create_expert_and_data_parallel(2)
    engine, _, _, _ = deepspeed_amd.initialize(model=model, config=config)
    torch.manual_seed(7 + rank)
    data = torch.randint(0, cfg.vocab_size, (2, 32))
    losses = []
    for _ in range(steps):
        loss = engine(data, labels=data)
        engine.backward(loss)
        engine.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0], losses
    engine.destroy()
    return losses


def test_mixtral_ep2_zero1():
    run_distributed(_mixtral_ep2, world_size=2)


def test_mixtral_generate_and_serving_cpu():
    """MoE decode path: KV-cache generate + continuous batching."""
    import torch
    from deepspeed_amd.inference.engine import InferenceEngine
    from deepspeed_amd.inference.serving import ContinuousBatchingEngine
    from deepspeed_amd.models.mixtral import (MIXTRAL_CONFIGS,
                                              MixtralForCausalLM)
    torch.manual_seed(0)
    model = MixtralForCausalLM(MIXTRAL_CONFIGS["mixtral-tiny"]).eval()
    eng = InferenceEngine(model)
    ids = torch.randint(0, model.cfg.vocab_size, (1, 7))
    out = eng.generate(ids, max_new_tokens=6)
    assert out.shape == (1, 13)
    # NOTE: capacity-based gating sees different token counts in decode
    # vs full recompute, so exact logits equality is not expected for
    # MoE; determinism between the two CACHED paths is.
    cb = ContinuousBatchingEngine(model, max_batch=2)
    rid = cb.add_request(ids[0], max_new_tokens=6)
    res = cb.run()
    assert torch.equal(res[rid], out[0])
