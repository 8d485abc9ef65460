"""ZenFlow selective-offload optimizer (stage 1/2 variant)."""
import torch

from tests.common import run_distributed


def _zenflow_train(overlap=False):
    import deepspeed_amd as ds
    torch.manual_seed(0)
    model = torch.nn.Sequential(torch.nn.Linear(64, 64), torch.nn.Tanh(),
                                torch.nn.Linear(64, 8))
    cfg = {
        "train_micro_batch_size_per_gpu": 4,
        "optimizer": {"type": "AdamW", "params": {"lr": 5e-3}},
        "bf16": {"enabled": True},
        "zero_optimization": {
            "stage": 2,
            "zenflow": {"topk_ratio": 0.05, "update_interval": 4,
                        "select_interval": 8, "overlap_step": overlap},
        },
    }
    engine, opt, _, _ = ds.initialize(model=model, config=cfg)
    from deepspeed_amd.runtime.zenflow import ZenFlowZeroOptimizer
    assert isinstance(engine.optimizer, ZenFlowZeroOptimizer)
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    x = torch.randn(4, 64, device=dev).bfloat16()
    y = torch.randn(4, 8, device=dev).bfloat16()
    losses = []
    shard_before_lazy = None
    zf = engine.optimizer
    for i in range(8):
        out = engine(x)
        loss = (out - y).float().pow(2).mean()
        engine.backward(loss)
        engine.step()
        losses.append(loss.item())
        if i == 1:
            # between lazy boundaries only hot channels move
            shard_before_lazy = zf.buckets[0].shard16.clone()
        if i == 2 and shard_before_lazy is not None:
            changed = (zf.buckets[0].shard16 !=
                       shard_before_lazy).float().mean().item()
            # only ~topk_ratio of entries should have changed
            assert changed < 0.25, f"too many channels moved: {changed}"
    # training makes progress
    assert losses[-1] < losses[0], losses
    # host masters exist and are finite
    st = zf._zf[0]
    assert torch.isfinite(st["master_cpu"]).all()
    # state roundtrip
    sd = zf.state_dict()
    zf.load_state_dict(sd)
    return losses[-1]


def test_zenflow_stage2_cpu():
    run_distributed(_zenflow_train, world_size=1)


def test_zenflow_world2():
    run_distributed(_zenflow_train, world_size=2)


def test_zenflow_overlap_step():
    """Async host-Adam worker: converges, threads drained at ckpt."""
    run_distributed(_zenflow_train, world_size=1, kwargs={"overlap": True})


import pytest  # noqa: E402


@pytest.mark.gpu
def test_zenflow_gpu_single():
    """Pinned-host masters + hot-set GPU updates on a real device."""
    run_distributed(_zenflow_train, world_size=1, backend="nccl")


def _zf3_body(steps=16):
    """ZenFlow over ZeRO-3: hot channels update immediately, cold mass
    lazily; converges close to the plain stage-3 run."""
    import torch
    import torch.distributed as tdist
    import deepspeed_amd
    from deepspeed_amd.comm import groups
    from deepspeed_amd.runtime.zenflow import ZenFlowZeroStage3Optimizer
    from tests.simple_model import SimpleModel, make_batches
    rank = tdist.get_rank()
    world = tdist.get_world_size()

    def run(zf):
        groups.reset_groups()
        torch.manual_seed(11)
        model = SimpleModel(32)
        zcfg = {"stage": 3}
        if zf:
            zcfg["zenflow"] = {"topk_ratio": 0.5, "update_interval": 2,
                               "select_interval": 4}
        config = {
            "train_micro_batch_size_per_gpu": 4,
            "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}},
            "zero_optimization": zcfg,
            "bf16": {"enabled": True},
        }
        engine, _, _, _ = deepspeed_amd.initialize(model=model,
                                                   config=config)
        if zf:
            assert isinstance(engine.optimizer, ZenFlowZeroStage3Optimizer)
        batches = make_batches(world, 4, 32, seed=5,
                               dtype=torch.bfloat16)
        x, y = batches[rank]  # fixed batch: memorization measures the
        losses = []           # optimizer, not generalization noise
        for i in range(steps):
            loss = engine(x, y)
            engine.backward(loss)
            engine.step()
            losses.append(loss.item())
        return losses

    l_zf = run(True)
    l_plain = run(False)
    # bounded-staleness scheme at toy scale is noisy step-to-step:
    # compare the tail mean against the head
    import statistics
    assert statistics.mean(l_zf[-4:]) < statistics.mean(l_zf[:4]), \
        f"zenflow3 no progress: {l_zf}"
    assert abs(statistics.mean(l_zf[-4:]) -
               statistics.mean(l_plain[-4:])) < 0.5, (l_zf, l_plain)
    return True


def test_zenflow_stage3_world2():
    from tests.common import run_distributed
    run_distributed(_zf3_body, world_size=2)


def test_zenflow_stage3_overlap_thread():
    from tests.common import run_distributed

    def wrap():
        return _zf3_overlap()
    run_distributed(_zf3_overlap, world_size=1)


def _zf3_overlap(steps=14):
    import torch
    import deepspeed_amd
    from deepspeed_amd.comm import groups
    from tests.simple_model import SimpleModel, make_batches
    groups.reset_groups()
    torch.manual_seed(11)
    model = SimpleModel(32)
    config = {
        "train_micro_batch_size_per_gpu": 4,
        "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}},
        "zero_optimization": {"stage": 3,
                              "zenflow": {"topk_ratio": 0.2,
                                          "update_interval": 2,
                                          "overlap_step": True}},
        "bf16": {"enabled": True},
    }
    engine, _, _, _ = deepspeed_amd.initialize(model=model, config=config)
    x, y = make_batches(1, 4, 32, dtype=torch.bfloat16)[0]
    losses = []
    for _ in range(steps):
        loss = engine(x, y)
        engine.backward(loss)
        engine.step()
        losses.append(loss.item())
    engine.optimizer._drain_threads()
    assert losses[-1] < losses[0] * 0.9, losses
    return True
