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


def _rollout_body():
    """Rollout interface: left-padded mixed-length prompts bucket by true
    length, n samples per prompt, right-padded RolloutBatch out."""
    import torch
    import deepspeed_amd
    from deepspeed_amd.comm import groups
    from deepspeed_amd.models.llama import LLAMA_CONFIGS, LlamaForCausalLM
    from deepspeed_amd.runtime.rollout import (
        HybridEngineRollout, RolloutBatch, RolloutRequest, SamplingConfig,
        get_rollout_engine)
    groups.reset_groups()
    torch.manual_seed(0)
    cfg = LLAMA_CONFIGS["llama-tiny"]
    model = LlamaForCausalLM(cfg)
    config = {
        "train_micro_batch_size_per_gpu": 3,
        "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}},
        "zero_optimization": {"stage": 3},
        "bf16": {"enabled": True},
    }
    engine, _, _, _ = deepspeed_amd.initialize(model=model, config=config)
    ro = get_rollout_engine(engine, pad_token_id=0)
    assert isinstance(ro, HybridEngineRollout)
    # three prompts, true lengths 8 / 8 / 5, left-padded to 8
    ids = torch.randint(1, cfg.vocab_size, (3, 8))
    mask = torch.ones(3, 8, dtype=torch.long)
    ids[2, :3] = 0
    mask[2, :3] = 0
    req = RolloutRequest(prompt_ids=ids, prompt_attention_mask=mask)
    samp = SamplingConfig(max_new_tokens=6, temperature=0.8, top_p=0.9,
                          top_k=20, n_samples_per_prompt=2)
    batch = ro.generate(req, samp)
    assert isinstance(batch, RolloutBatch)
    assert batch.batch_size == 6
    assert batch.seq_len == 8 + 6
    assert batch.response_start_idx.tolist() == [8, 8, 8, 8, 5, 5]
    # samples of one prompt share its prompt tokens
    assert torch.equal(batch.input_ids[0, :8], batch.input_ids[1, :8])
    assert torch.equal(batch.input_ids[0, :8], ids[0])
    # short prompt: unpadded tokens then response, then right pad
    assert torch.equal(batch.input_ids[4, :5], ids[2, 3:])
    assert batch.attention_mask[4, :11].all()
    assert (batch.attention_mask[4, 11:] == 0).all()
    ro.sync_weights(0)  # no-op: co-located
    ro.shutdown()
    # greedy determinism: temperature 0 twice gives identical rollouts
    g = SamplingConfig(max_new_tokens=4, temperature=0.0)
    b1 = ro.generate(req, g)
    b2 = ro.generate(req, g)
    assert torch.equal(b1.input_ids, b2.input_ids)
    return True


def test_rollout_interface():
    from tests.common import run_distributed
    run_distributed(_rollout_body, world_size=1)


def test_hybrid_hipgraph_plumbing_cpu():
    """use_hipgraph validates greedy-only and needs a GPU (the decode
    core itself is GPU-covered by test_engine_gpu hipgraph tests)."""
    import pytest
    import torch
    import deepspeed_amd
    from deepspeed_amd.comm import groups
    from deepspeed_amd.models.llama import LLAMA_CONFIGS, LlamaForCausalLM
    groups.reset_groups()
    torch.manual_seed(0)
    model = LlamaForCausalLM(LLAMA_CONFIGS["llama-tiny"])
    config = {"train_micro_batch_size_per_gpu": 2,
              "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}},
              "bf16": {"enabled": True},
              "zero_optimization": {"stage": 0}}
    import torch.distributed as tdist
    if not tdist.is_initialized():
        tdist.init_process_group(
            "gloo", init_method="tcp://127.0.0.1:29631",
            rank=0, world_size=1)
    engine, _, _, _ = deepspeed_amd.initialize(model=model, config=config)
    ids = torch.randint(0, 512, (1, 8))
    with pytest.raises(ValueError, match="greedy-only"):
        engine.generate(ids, max_new_tokens=2, temperature=0.5,
                        use_hipgraph=True)
    if not torch.cuda.is_available():
        with pytest.raises(AssertionError, match="needs a GPU"):
            engine.generate(ids, max_new_tokens=2, use_hipgraph=True)


import pytest  # noqa: E402


@pytest.mark.gpu
def test_hybrid_hipgraph_matches_eager_gpu():
    """Graph-captured rollout decode off gathered ZeRO-3 shards matches
    the eager hybrid generate token-for-token."""
    run_distributed(_hybrid_hipgraph_gpu_body, world_size=1,
                    backend="nccl")


def _hybrid_hipgraph_gpu_body():
    import torch
    import deepspeed_amd
    from deepspeed_amd.comm import groups
    from deepspeed_amd.models.llama import LLAMA_CONFIGS, LlamaForCausalLM
    groups.reset_groups()
    torch.manual_seed(0)
    model = LlamaForCausalLM(LLAMA_CONFIGS["llama-tiny"])
    config = {"train_micro_batch_size_per_gpu": 2,
              "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}},
              "bf16": {"enabled": True},
              "zero_optimization": {"stage": 3}}
    engine, _, _, _ = deepspeed_amd.initialize(model=model, config=config)
    ids = torch.randint(0, 512, (2, 8), device="cuda")
    loss = engine(ids, labels=ids)
    engine.backward(loss)
    engine.step()
    eager = engine.generate(ids, max_new_tokens=8)
    graphed = engine.generate(ids, max_new_tokens=8, use_hipgraph=True)
    assert torch.equal(eager, graphed), (eager, graphed)
    # training continues after graph capture/replay
    loss = engine(ids, labels=ids)
    engine.backward(loss)
    engine.step()
    assert torch.isfinite(torch.tensor(loss.item()))
    return True


def test_rollout_dataclass_validation():
    """RolloutRequest/RolloutBatch reject malformed shapes (ref
    runtime/rollout/base.py __post_init__ checks)."""
    import pytest
    import torch
    from deepspeed_amd.runtime.rollout import RolloutBatch, RolloutRequest
    ids = torch.zeros(2, 5, dtype=torch.long)
    with pytest.raises(ValueError):
        RolloutRequest(prompt_ids=torch.zeros(5, dtype=torch.long),
                       prompt_attention_mask=torch.ones(5))
    with pytest.raises(ValueError):
        RolloutRequest(prompt_ids=ids,
                       prompt_attention_mask=torch.ones(2, 4))
    b = RolloutBatch(input_ids=ids, attention_mask=torch.ones(2, 5),
                     response_start_idx=torch.tensor([3, 2]))
    assert b.batch_size == 2 and b.seq_len == 5
    with pytest.raises(ValueError):
        RolloutBatch(input_ids=ids, attention_mask=torch.ones(2, 5),
                     response_start_idx=torch.tensor([3]))
