"""Hybrid engine: ZeRO-3 train -> generate -> train round trip."""
import torch

from tests.common import run_distributed


def _train_generate_train():
    import torch.distributed as tdist
    import deepspeed_amd
    from deepspeed_amd.comm import groups
    from deepspeed_amd.models.llama import LLAMA_CONFIGS, LlamaForCausalLM
    groups.reset_groups()
    cfg = LLAMA_CONFIGS["llama-tiny"]
    torch.manual_seed(0)
    model = LlamaForCausalLM(cfg)
    config = {
        "train_micro_batch_size_per_gpu": 2,
        "optimizer": {"type": "AdamW", "params": {"lr": 3e-4}},
        "zero_optimization": {"stage": 3},
        "bf16": {"enabled": True},
    }
    engine, _, _, _ = deepspeed_amd.initialize(model=model, config=config)
    data = torch.randint(0, cfg.vocab_size, (2, 32))

    def train_step():
        loss = engine(data, labels=data)
        engine.backward(loss)
        engine.step()
        return loss.item()

    l0 = train_step()
    out = engine.generate(data[:, :8], max_new_tokens=8)
    assert out.shape == (2, 16)
    # training continues normally after generation
    l1 = train_step()
    l2 = train_step()
    assert l2 < l0, (l0, l1, l2)
    return True


def test_hybrid_engine_2rank():
    assert all(run_distributed(_train_generate_train, world_size=2))


def _hybrid_offload_body():
    """RLHF rollout with optimizer states pushed to host: weights gather,
    generation runs, states reload, training resumes."""
    import torch
    import deepspeed_amd
    from deepspeed_amd.comm import groups
    from deepspeed_amd.models.llama import LLAMA_CONFIGS, LlamaForCausalLM
    from deepspeed_amd.runtime.hybrid_engine import generate
    groups.reset_groups()
    torch.manual_seed(0)
    cfg = LLAMA_CONFIGS["llama-tiny"]
    model = LlamaForCausalLM(cfg)
    config = {
        "train_micro_batch_size_per_gpu": 2,
        "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}},
        "zero_optimization": {"stage": 3},
        "bf16": {"enabled": True},
    }
    engine, _, _, _ = deepspeed_amd.initialize(model=model, config=config)
    data = torch.randint(0, cfg.vocab_size, (2, 32))
    loss = engine(data, labels=data)
    engine.backward(loss)
    engine.step()
    out = generate(engine, data[:, :8], max_new_tokens=4,
                   offload_states_during_generate=True)
    assert out.shape[1] == 12
    for sg in engine.optimizer.sub_groups:  # states back on train device
        assert sg.master32.device == engine.optimizer.master32_device \
            if hasattr(engine.optimizer, "master32_device") else True
    loss2 = engine(data, labels=data)
    engine.backward(loss2)
    engine.step()
    assert torch.isfinite(torch.tensor(loss2.item()))
    return True


def test_hybrid_generate_with_state_offload():
    from tests.common import run_distributed
    run_distributed(_hybrid_offload_body, world_size=1)
