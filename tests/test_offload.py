"""ZeRO-Offload (CPU optimizer) numerics; native cpu_adam_step vs AdamW."""
import torch

import deepspeed_amd
from tests.common import run_distributed
from tests.simple_model import SimpleModel, make_batches, reference_adamw_training

HIDDEN = 32
LR = 1e-3


def test_cpu_adam_step_matches_torch():
    from deepspeed_amd.ops.loader import get_ext, has_ext
    if not has_ext():
        import pytest
        pytest.skip("extension not built")
    ext = get_ext()
    torch.manual_seed(0)
    n = 12345
    p1 = torch.randn(n)
    p2 = p1.clone().requires_grad_(True)
    m = torch.zeros(n)
    v = torch.zeros(n)
    opt = torch.optim.AdamW([p2], lr=1e-2, weight_decay=0.01)
    for step in range(1, 5):
        g = torch.randn(n)
        ext.cpu_adam_step(p1, g, m, v, None, 1e-2, 0.9, 0.999, 1e-8, step,
                          1, 1, 0.01, 1.0)
        p2.grad = g.clone()
        opt.step()
    assert torch.allclose(p1, p2.detach(), atol=1e-6), (p1 - p2).abs().max()


def test_cpu_adam_optimizer_matches_fused():
    from deepspeed_amd.ops.cpu_adam import DeepSpeedCPUAdam
    from deepspeed_amd.ops.adam import FusedAdam
    torch.manual_seed(0)
    p1 = torch.randn(500, requires_grad=True)
    p2 = p1.detach().clone().requires_grad_(True)
    o1 = DeepSpeedCPUAdam([p1], lr=1e-2, weight_decay=0.01)
    o2 = FusedAdam([p2], lr=1e-2, weight_decay=0.01)
    for _ in range(4):
        g = torch.randn(500)
        p1.grad = g.clone()
        p2.grad = g.clone()
        o1.step()
        o2.step()
    assert torch.allclose(p1, p2, atol=1e-5)


def _zero3_offload_train(steps=4, device="cpu"):
    import torch.distributed as tdist
    from deepspeed_amd.comm import groups
    groups.reset_groups()
    rank = tdist.get_rank()
    world = tdist.get_world_size()
    torch.manual_seed(11)
    model = SimpleModel(HIDDEN)
    config = {
        "train_micro_batch_size_per_gpu": 4,
        "optimizer": {"type": "AdamW", "params": {"lr": LR}},
        "zero_optimization": {"stage": 3, "reduce_bucket_size": 2000,
                              "sub_group_size": 1500,
                              "offload_optimizer": {"device": device,
                                                    "nvme_path":
                                                    "/tmp/dsamd_test_swap"}},
        "bf16": {"enabled": True},
    }
    engine, opt, _, _ = deepspeed_amd.initialize(model=model, config=config)
    if device == "cpu":
        from deepspeed_amd.ops.cpu_adam import DeepSpeedCPUAdam
        assert isinstance(engine.optimizer.optimizer, DeepSpeedCPUAdam)
    else:
        assert engine.optimizer.nvme_swapper is not None
    batches = make_batches(steps * world, 4, HIDDEN, dtype=torch.bfloat16)
    for i in range(steps):
        x, y = batches[i * world + rank]
        loss = engine(x, y)
        engine.backward(loss)
        engine.step()
    from deepspeed_amd.runtime.zero.stage3_params import (all_gather_params,
                                                          ZeroParamStatus)
    params = list(model.parameters())
    need = [p for p in params
            if p.ds_status == ZeroParamStatus.NOT_AVAILABLE]
    all_gather_params(need, None, async_op=False).wait()
    return [p.detach().float().cpu() for p in params]


def test_zero3_cpu_offload_matches_reference():
    steps, world = 4, 2
    results = run_distributed(_zero3_offload_train, world_size=world,
                              args=(steps,))
    batches = make_batches(steps * world, 4, HIDDEN)
    merged = []
    for i in range(steps):
        xs = torch.cat([batches[i * world + r][0] for r in range(world)])
        ys = torch.cat([batches[i * world + r][1] for r in range(world)])
        merged.append((xs, ys))
    ref_model = reference_adamw_training(lambda: SimpleModel(HIDDEN), merged,
                                         lr=LR)
    ref = [p.detach().float() for p in ref_model.parameters()]
    for g, e in zip(results[0], ref):
        assert torch.allclose(g, e, atol=3e-2, rtol=3e-2), \
            (g - e).abs().max()


def test_zero3_nvme_offload_matches_reference():
    steps, world = 4, 2
    results = run_distributed(_zero3_offload_train, world_size=world,
                              args=(steps, "nvme"))
    batches = make_batches(steps * world, 4, HIDDEN)
    merged = []
    for i in range(steps):
        xs = torch.cat([batches[i * world + r][0] for r in range(world)])
        ys = torch.cat([batches[i * world + r][1] for r in range(world)])
        merged.append((xs, ys))
    ref_model = reference_adamw_training(lambda: SimpleModel(HIDDEN), merged,
                                         lr=LR)
    ref = [p.detach().float() for p in ref_model.parameters()]
    for g, e in zip(results[0], ref):
        assert torch.allclose(g, e, atol=3e-2, rtol=3e-2), \
            (g - e).abs().max()


def _nvme_ckpt_roundtrip():
    import tempfile
    import deepspeed_amd as ds
    from tests.simple_model import SimpleModel
    torch.manual_seed(0)
    nvme_dir = tempfile.mkdtemp(prefix="nvme_swap")
    cfg = {
        "train_micro_batch_size_per_gpu": 4,
        "optimizer": {"type": "AdamW", "params": {"lr": 5e-3}},
        "bf16": {"enabled": True},
        "zero_optimization": {
            "stage": 3, "sub_group_size": 200,
            "offload_optimizer": {"device": "nvme",
                                  "nvme_path": nvme_dir},
        },
    }
    engine, _, _, _ = ds.initialize(model=SimpleModel(32), config=cfg)
    x = torch.randn(4, 32).bfloat16()
    y = torch.randn(4, 32).bfloat16()
    for _ in range(3):
        loss = engine(x, y)
        engine.backward(loss)
        engine.step()
    sd = engine.optimizer.state_dict()
    masters = [m.clone() for m in sd["fp32_flat_groups"]]
    assert "nvme_exp_avg" in sd and len(sd["nvme_exp_avg"]) == len(masters)
    assert any(m.abs().sum() > 0 for m in masters)
    # perturb on-disk state by stepping more, then restore
    for _ in range(2):
        loss = engine(x, y)
        engine.backward(loss)
        engine.step()
    engine.optimizer.load_state_dict(sd)
    sd2 = engine.optimizer.state_dict()
    for a, b in zip(masters, sd2["fp32_flat_groups"]):
        assert torch.allclose(a, b, atol=1e-6)


def test_nvme_state_checkpoint_roundtrip():
    run_distributed(_nvme_ckpt_roundtrip, world_size=1)


# ---------------------------------------------------- ZeRO-Infinity params

def _zero3_param_offload_train(steps=4, pdevice="cpu", odevice="cpu",
                               max_in_cpu=int(1e9)):
    """offload_param: 16-bit shard slabs in host RAM (cpu) or NVMe-backed
    with an LRU host budget (nvme). Must train to the same weights as the
    non-offloaded reference."""
    import torch.distributed as tdist
    from deepspeed_amd.comm import groups
    groups.reset_groups()
    rank = tdist.get_rank()
    world = tdist.get_world_size()
    torch.manual_seed(11)
    model = SimpleModel(HIDDEN)
    config = {
        "train_micro_batch_size_per_gpu": 4,
        "optimizer": {"type": "AdamW", "params": {"lr": LR}},
        "zero_optimization": {
            "stage": 3, "reduce_bucket_size": 2000,
            "sub_group_size": 800,  # several sub-groups => real eviction
            "stage3_param_persistence_threshold": 0,
            "offload_param": {"device": pdevice,
                              "nvme_path": "/tmp/dsamd_test_pswap",
                              "max_in_cpu": max_in_cpu},
            "offload_optimizer": {"device": odevice,
                                  "nvme_path": "/tmp/dsamd_test_pswap"}},
        "bf16": {"enabled": True},
    }
    engine, opt, _, _ = deepspeed_amd.initialize(model=model, config=config)
    zopt = engine.optimizer
    assert zopt.param_offload
    for sg in zopt.sub_groups:
        assert sg.flat16 is None or not sg.flat16.is_cuda \
            or not torch.cuda.is_available(), \
            "param slab should be host-resident (or NVMe-evicted)"
    if pdevice == "nvme":
        assert zopt.param_swapper is not None
    batches = make_batches(steps * world, 4, HIDDEN, dtype=torch.bfloat16)
    for i in range(steps):
        x, y = batches[i * world + rank]
        loss = engine(x, y)
        engine.backward(loss)
        engine.step()
    evicted = 0
    if pdevice == "nvme":
        sw = zopt.param_swapper
        evicted = len(sw._on_disk)
        assert sw.resident_elems() <= sw.max_in_cpu or \
            len(sw._resident) <= 1
    from deepspeed_amd.runtime.zero.stage3_params import (all_gather_params,
                                                          ZeroParamStatus)
    # gather via the optimizer path so evicted slabs swap back in
    params = list(model.parameters())
    need = [p for p in params
            if p.ds_status == ZeroParamStatus.NOT_AVAILABLE]
    if need:
        zopt._gather_grouped(need, async_op=False).wait()
    return [p.detach().float().cpu() for p in params], evicted


def _check_param_offload(pdevice, odevice, max_in_cpu=int(1e9),
                         expect_evictions=False):
    steps, world = 4, 2
    results = run_distributed(_zero3_param_offload_train, world_size=world,
                              args=(steps, pdevice, odevice, max_in_cpu))
    batches = make_batches(steps * world, 4, HIDDEN)
    merged = []
    for i in range(steps):
        xs = torch.cat([batches[i * world + r][0] for r in range(world)])
        ys = torch.cat([batches[i * world + r][1] for r in range(world)])
        merged.append((xs, ys))
    ref_model = reference_adamw_training(lambda: SimpleModel(HIDDEN), merged,
                                         lr=LR)
    ref = [p.detach().float() for p in ref_model.parameters()]
    got, evicted = results[0]
    for g, e in zip(got, ref):
        assert torch.allclose(g, e, atol=3e-2, rtol=3e-2), \
            (g - e).abs().max()
    if expect_evictions:
        assert evicted > 0, "no slab was ever written to NVMe"


def test_zero3_param_offload_cpu():
    _check_param_offload("cpu", "cpu")


def test_zero3_param_offload_nvme_eviction():
    # budget below total param elements => slabs must spill + reload
    _check_param_offload("nvme", "cpu", max_in_cpu=900,
                         expect_evictions=True)


def test_zero3_param_offload_nvme_full_infinity():
    # params on NVMe + optimizer state on NVMe: the full Infinity tier
    _check_param_offload("nvme", "nvme", max_in_cpu=900,
                         expect_evictions=True)


def _param_offload_ckpt(tmpdir):
    import torch.distributed as tdist
    from deepspeed_amd.comm import groups
    groups.reset_groups()
    torch.manual_seed(11)
    model = SimpleModel(HIDDEN)
    config = {
        "train_micro_batch_size_per_gpu": 4,
        "optimizer": {"type": "AdamW", "params": {"lr": LR}},
        "zero_optimization": {
            "stage": 3, "sub_group_size": 800,
            "stage3_param_persistence_threshold": 0,
            "offload_param": {"device": "nvme",
                              "nvme_path": "/tmp/dsamd_test_pswap2",
                              "max_in_cpu": 900},
            "offload_optimizer": {"device": "cpu"}},
        "bf16": {"enabled": True},
    }
    engine, _, _, _ = deepspeed_amd.initialize(model=model, config=config)
    batches = make_batches(6, 4, HIDDEN, dtype=torch.bfloat16)
    for x, y in batches[:3]:
        loss = engine(x, y)
        engine.backward(loss)
        engine.step()
    engine.save_checkpoint(tmpdir, tag="ck")
    before = [sg.master32.detach().clone()
              for sg in engine.optimizer.sub_groups]

    groups.reset_groups()
    torch.manual_seed(11)
    model2 = SimpleModel(HIDDEN)
    engine2, _, _, _ = deepspeed_amd.initialize(model=model2, config=config)
    engine2.load_checkpoint(tmpdir, tag="ck")
    after = [sg.master32.detach().clone()
             for sg in engine2.optimizer.sub_groups]
    for b, a in zip(before, after):
        assert torch.allclose(b, a, atol=1e-6), (b - a).abs().max()
    # resumed training still works (slabs resident/evicted correctly)
    x, y = batches[3]
    loss = engine2(x, y)
    engine2.backward(loss)
    engine2.step()
    return True


def test_param_offload_checkpoint_roundtrip(tmp_path):
    run_distributed(_param_offload_ckpt, world_size=1,
                    args=(str(tmp_path),))


def test_param_offload_checkpoint_roundtrip_world2(tmp_path):
    run_distributed(_param_offload_ckpt, world_size=2,
                    args=(str(tmp_path),))


def _partial_offload_body(steps=4):
    """offload_optimizer.ratio: a fraction of sub-groups keeps GPU-fused
    state while the rest offloads; trains to the same result."""
    import torch
    import torch.distributed as tdist
    import deepspeed_amd
    from deepspeed_amd.comm import groups
    from tests.simple_model import SimpleModel, make_batches
    rank = tdist.get_rank()
    world = tdist.get_world_size()

    def run(ratio):
        groups.reset_groups()
        torch.manual_seed(11)
        model = SimpleModel(HIDDEN)
        config = {
            "train_micro_batch_size_per_gpu": 4,
            "optimizer": {"type": "AdamW", "params": {"lr": LR}},
            "zero_optimization": {
                "stage": 3, "sub_group_size": 800,
                "offload_optimizer": {"device": "cpu", "ratio": ratio}},
            "bf16": {"enabled": True},
        }
        engine, _, _, _ = deepspeed_amd.initialize(model=model,
                                                   config=config)
        zopt = engine.optimizer
        if 0 < ratio < 1:
            kinds = {sg.offload for sg in zopt.sub_groups}
            assert kinds == {True, False}, \
                f"expected mixed placement, got {kinds}"
        batches = make_batches(steps * world, 4, HIDDEN,
                               dtype=torch.bfloat16)
        for i in range(steps):
            x, y = batches[i * world + rank]
            loss = engine(x, y)
            engine.backward(loss)
            engine.step()
        shards = [sg.master32.detach().cpu().clone()
                  for sg in zopt.sub_groups]
        engine.optimizer.destroy()
        return shards

    s_half = run(0.5)
    s_full = run(1.0)
    for a, b in zip(s_half, s_full):
        err = (a - b).abs().max().item()
        assert err < 1e-4, f"partial offload diverged: {err}"
    return True


def test_zero3_partial_offload_ratio():
    run_distributed(_partial_offload_body, world_size=2)
