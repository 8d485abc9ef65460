"""Fragment APIs under ZeRO stages 2 and 3 (2-rank)."""
import torch

from tests.common import run_distributed
from tests.simple_model import SimpleModel, make_batches

HIDDEN = 32


def _run(stage):
    import torch.distributed as tdist
    import deepspeed_amd
    from deepspeed_amd.comm import groups
    from deepspeed_amd.utils.tensor_fragment import (
        safe_get_full_fp32_param, safe_get_full_grad,
        safe_get_full_optimizer_state, safe_set_full_fp32_param)
    groups.reset_groups()
    rank = tdist.get_rank()
    torch.manual_seed(11)
    model = SimpleModel(HIDDEN)
    config = {
        "train_micro_batch_size_per_gpu": 4,
        "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}},
        "zero_optimization": {"stage": stage, "reduce_bucket_size": 2000,
                              "sub_group_size": 1500},
        "bf16": {"enabled": True},
    }
    engine, _, _, _ = deepspeed_amd.initialize(model=model, config=config)
    batches = make_batches(4, 4, HIDDEN, dtype=torch.bfloat16)
    x, y = batches[rank]
    loss = engine(x, y)
    engine.backward(loss)

    p = next(iter(model.parameters()))
    shape = p.ds_shape if hasattr(p, "ds_shape") else p.shape
    g = safe_get_full_grad(p, engine.optimizer)
    assert g is not None and tuple(g.shape) == tuple(shape)
    assert g.abs().sum() > 0

    engine.step()
    full = safe_get_full_fp32_param(p, engine.optimizer)
    assert full is not None and tuple(full.shape) == tuple(shape)
    m = safe_get_full_optimizer_state(p, "exp_avg", engine.optimizer)
    assert m is not None and m.abs().sum() > 0

    # round-trip set
    newv = torch.full(tuple(shape), 0.5)
    safe_set_full_fp32_param(p, newv, engine.optimizer)
    back = safe_get_full_fp32_param(p, engine.optimizer)
    assert torch.allclose(back, newv, atol=1e-6), (back - newv).abs().max()
    engine.destroy()
    return True


def test_fragment_api_stage2():
    assert all(run_distributed(_run, world_size=2, args=(2,)))


def test_fragment_api_stage3():
    assert all(run_distributed(_run, world_size=2, args=(3,)))
