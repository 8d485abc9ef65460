"""Checkpoint round-trips: save/load resume, zero_to_fp32, universal."""
import os
import tempfile

import torch

from tests.common import run_distributed
from tests.simple_model import SimpleModel, make_batches

HIDDEN = 32
LR = 1e-3


def _make_engine(stage, tmpdir=None):
    import deepspeed_amd
    torch.manual_seed(11)
    model = SimpleModel(HIDDEN)
    config = {
        "train_micro_batch_size_per_gpu": 4,
        "optimizer": {"type": "AdamW", "params": {"lr": LR}},
        "zero_optimization": {"stage": stage, "reduce_bucket_size": 2000,
                              "sub_group_size": 1500},
        "bf16": {"enabled": True},
    }
    engine, _, _, _ = deepspeed_amd.initialize(model=model, config=config)
    return engine


def _train_some(engine, batches, n):
    for i in range(n):
        x, y = batches[i]
        loss = engine(x, y)
        engine.backward(loss)
        engine.step()
    return loss.item()


def _roundtrip(stage, ckpt_dir):
    import torch.distributed as tdist
    from deepspeed_amd.comm import groups
    groups.reset_groups()
    rank = tdist.get_rank()
    world = tdist.get_world_size()
    batches = make_batches(20, 4, HIDDEN, dtype=torch.bfloat16,
                           seed=31 + rank)
    e1 = _make_engine(stage)
    _train_some(e1, batches, 4)
    e1.save_checkpoint(ckpt_dir)
    # continue 3 more steps -> reference trajectory
    ref_loss = _train_some(e1, batches[4:], 3)
    e1.destroy()

    # fresh engine, load, re-run the same 3 steps
    e2 = _make_engine(stage)
    e2.load_checkpoint(ckpt_dir)
    got_loss = _train_some(e2, batches[4:], 3)
    e2.destroy()
    assert abs(ref_loss - got_loss) < 1e-5, (ref_loss, got_loss)
    return True


def test_checkpoint_roundtrip_stage2():
    with tempfile.TemporaryDirectory() as d:
        assert all(run_distributed(_roundtrip, world_size=2, args=(2, d)))


def test_checkpoint_roundtrip_stage3():
    with tempfile.TemporaryDirectory() as d:
        assert all(run_distributed(_roundtrip, world_size=2, args=(3, d)))


def _save_and_export(stage, ckpt_dir):
    import torch.distributed as tdist
    from deepspeed_amd.comm import groups
    groups.reset_groups()
    rank = tdist.get_rank()
    batches = make_batches(8, 4, HIDDEN, dtype=torch.bfloat16,
                           seed=31 + rank)
    e = _make_engine(stage)
    _train_some(e, batches, 3)
    e.save_checkpoint(ckpt_dir)
    # return the full bf16 params for comparison
    if stage == 3:
        from deepspeed_amd.runtime.zero.stage3_params import (
            all_gather_params, ZeroParamStatus)
        params = list(e.module.parameters())
        need = [p for p in params
                if p.ds_status == ZeroParamStatus.NOT_AVAILABLE]
        all_gather_params(need, None, async_op=False).wait()
    named = {n: p.detach().float().clone()
             for n, p in e.module.named_parameters()}
    e.destroy()
    return named


def _check_fp32_export(stage):
    with tempfile.TemporaryDirectory() as d:
        results = run_distributed(_save_and_export, world_size=2,
                                  args=(stage, d))
        from deepspeed_amd.utils.zero_to_fp32 import \
            get_fp32_state_dict_from_zero_checkpoint
        sd = get_fp32_state_dict_from_zero_checkpoint(d)
        ref = results[0]
        for name, expected in ref.items():
            assert name in sd, f"{name} missing from fp32 export"
            # fp32 master vs bf16 params: master is the precise one
            assert torch.allclose(sd[name].float(), expected, atol=1e-2), \
                f"{name}: {(sd[name].float() - expected).abs().max()}"


def test_zero_to_fp32_stage2():
    _check_fp32_export(2)


def test_zero_to_fp32_stage3():
    _check_fp32_export(3)


def _universal_load(stage, ckpt_dir, universal_dir):
    """Runs at world_size=1: loads a world-2 checkpoint via universal."""
    import torch.distributed as tdist
    from deepspeed_amd.comm import groups
    groups.reset_groups()
    e = _make_engine(stage)
    e.load_universal_checkpoint(universal_dir)
    named = {n: (p.detach().float().clone() if not hasattr(p, "ds_tensor")
                 else None) for n, p in e.module.named_parameters()}
    if stage == 3:
        from deepspeed_amd.runtime.zero.stage3_params import (
            all_gather_params, ZeroParamStatus)
        params = list(e.module.parameters())
        need = [p for p in params
                if p.ds_status == ZeroParamStatus.NOT_AVAILABLE]
        all_gather_params(need, None, async_op=False).wait()
        named = {n: p.detach().float().clone()
                 for n, p in e.module.named_parameters()}
    e.destroy()
    return named


def _check_universal(stage):
    with tempfile.TemporaryDirectory() as d:
        ckpt = os.path.join(d, "ckpt")
        uni = os.path.join(d, "uni")
        results = run_distributed(_save_and_export, world_size=2,
                                  args=(stage, ckpt))
        from deepspeed_amd.checkpoint.universal import ds_to_universal
        names = ds_to_universal(ckpt, uni)
        assert len(names) > 0
        got = run_distributed(_universal_load, world_size=1,
                              args=(stage, ckpt, uni))[0]
        for name, expected in results[0].items():
            assert torch.allclose(got[name], expected, atol=1e-2), \
                f"{name}: {(got[name] - expected).abs().max()}"


def test_universal_checkpoint_stage2_world2_to_1():
    _check_universal(2)


def test_universal_checkpoint_stage3_world2_to_1():
    _check_universal(3)


def _save16(stage, outdir):
    import torch.distributed as tdist
    from deepspeed_amd.comm import groups
    groups.reset_groups()
    e = _make_engine(stage)
    batches = make_batches(4, 4, HIDDEN, dtype=torch.bfloat16)
    _train_some(e, batches, 2)
    e.save_16bit_model(outdir)
    e.destroy()
    return True


def test_save_16bit_model_stage3():
    with tempfile.TemporaryDirectory() as d:
        assert all(run_distributed(_save16, world_size=2, args=(3, d)))
        sd = torch.load(os.path.join(d, "pytorch_model.bin"),
                        map_location="cpu", weights_only=False)
        # full shapes restored
        assert all(v.numel() > 0 for v in sd.values())
        assert any("linears.0.weight" in k for k in sd)


def test_fast_checkpoint_engine_roundtrip():
    from deepspeed_amd.runtime.checkpoint_engine import (
        FastCheckpointEngine, TorchCheckpointEngine, make_checkpoint_engine)
    eng = make_checkpoint_engine("fast")
    assert isinstance(eng, FastCheckpointEngine)
    sd = {"module": {"w": torch.randn(17, 3),
                     "b16": torch.randn(8).to(torch.bfloat16)},
          "step": 7, "nested": [torch.arange(5), "tag", (1, 2)]}
    with tempfile.TemporaryDirectory() as d:
        path = os.path.join(d, "ckpt.meta")
        eng.save(sd, path)
        back = eng.load(path)
    assert torch.equal(back["module"]["w"], sd["module"]["w"])
    assert torch.equal(back["module"]["b16"], sd["module"]["b16"])
    assert back["step"] == 7
    assert torch.equal(back["nested"][0], sd["nested"][0])
    assert back["nested"][1] == "tag"
    assert back["nested"][2] == (1, 2)


def test_decoupled_checkpoint_engine(tmp_path):
    """Background-process writer: save returns immediately, commit
    blocks until durable, mutations after save don't corrupt the file."""
    import time
    import torch
    from deepspeed_amd.runtime.checkpoint_engine import (
        DecoupledCheckpointEngine)
    eng = DecoupledCheckpointEngine()
    t = torch.randn(256, 256)
    snap = t.clone()
    path = str(tmp_path / "state.pt")
    t0 = time.time()
    eng.save({"w": t, "step": 7}, path)
    submit_time = time.time() - t0
    t.add_(100.0)  # trainer mutates right after save
    eng.commit("tag")
    loaded = eng.load(path)
    assert torch.allclose(loaded["w"], snap)  # snapshot, not mutated
    assert loaded["step"] == 7
    assert submit_time < 5.0
    eng.close()


def _moe_engine(lr=1e-3):
    import torch
    import deepspeed_amd as ds
    from deepspeed_amd.moe.layer import MoE
    M = 16

    class Net(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.inp = torch.nn.Linear(M, M)
            self.moe = MoE(M, torch.nn.Linear(M, M), num_experts=4,
                           ep_size=2, k=1, capacity_factor=8.0)

        def forward(self, x):
            h, _, _ = self.moe(self.inp(x))
            return h

    cfg = {"train_micro_batch_size_per_gpu": 2,
           "optimizer": {"type": "AdamW", "params": {"lr": lr}},
           "bf16": {"enabled": True},
           "zero_optimization": {"stage": 2}}
    engine, _, _, _ = ds.initialize(model=Net(), config=cfg)
    return engine


def _moe_uni_save(ckpt_dir):
    import torch
    import torch.distributed as tdist
    from deepspeed_amd.comm import groups
    groups.reset_groups()
    rank = tdist.get_rank()
    torch.manual_seed(7 + rank)  # rank-distinct experts
    e = _moe_engine()
    x = torch.randn(2, 16).bfloat16()
    for _ in range(2):
        loss = e(x).float().pow(2).mean()
        e.backward(loss)
        e.step()
    e.save_checkpoint(ckpt_dir)
    named = {n: p.detach().float().clone()
             for n, p in e.module.named_parameters()}
    e.destroy()
    return named


def _moe_uni_load(ckpt_dir, universal_dir):
    import torch
    import torch.distributed as tdist
    from deepspeed_amd.comm import groups
    groups.reset_groups()
    rank = tdist.get_rank()
    torch.manual_seed(7 + rank)
    e = _moe_engine()
    # perturb away from the saved state, then restore via universal
    x = torch.randn(2, 16).bfloat16() + 1.0
    loss = e(x).float().pow(2).mean()
    e.backward(loss)
    e.step()
    e.load_universal_checkpoint(universal_dir)
    named = {n: p.detach().float().clone()
             for n, p in e.module.named_parameters()}
    e.destroy()
    return named


def test_universal_checkpoint_moe_expert_buckets():
    """Expert buckets convert per expert-DP group ("@ep<off>" entries)
    and load back rank-correct experts (ref checkpoint/autoep role)."""
    with tempfile.TemporaryDirectory() as d:
        ckpt = os.path.join(d, "ckpt")
        uni = os.path.join(d, "uni")
        saved = run_distributed(_moe_uni_save, world_size=2, args=(ckpt,))
        from deepspeed_amd.checkpoint.universal import ds_to_universal
        names = ds_to_universal(ckpt, uni)
        ep_entries = [n for n in names if "@ep" in n]
        assert ep_entries, names  # expert params stored qualified
        got = run_distributed(_moe_uni_load, world_size=2,
                              args=(ckpt, uni))
        for r in range(2):
            for name, expected in saved[r].items():
                assert torch.allclose(got[r][name], expected, atol=1e-2), \
                    f"rank{r} {name}: " \
                    f"{(got[r][name] - expected).abs().max()}"


def _moe3_net_engine():
    import torch
    import deepspeed_amd as ds
    from deepspeed_amd.moe.layer import MoE
    M = 16

    class Net(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.inp = torch.nn.Linear(M, M)
            self.moe = MoE(M, torch.nn.Linear(M, M), num_experts=4,
                           ep_size=2, k=1, capacity_factor=8.0)

        def forward(self, x):
            h, _, _ = self.moe(self.inp(x))
            return h

    cfg = {"train_micro_batch_size_per_gpu": 2,
           "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}},
           "bf16": {"enabled": True},
           "zero_optimization": {"stage": 3, "sub_group_size": 500}}
    engine, _, _, _ = ds.initialize(model=Net(), config=cfg)
    return engine


def _gather_named(e):
    import torch
    from deepspeed_amd.runtime.zero.stage3_params import (
        ZeroParamStatus, all_gather_params)
    need = [p for p in e.module.parameters()
            if p.ds_status == ZeroParamStatus.NOT_AVAILABLE]
    all_gather_params(need, None, async_op=False).wait()
    return {n: p.detach().float().clone()
            for n, p in e.module.named_parameters()}


def _moe3_save(ckpt):
    import torch
    import torch.distributed as tdist
    from deepspeed_amd.comm import groups
    groups.reset_groups()
    torch.manual_seed(7 + tdist.get_rank())
    e = _moe3_net_engine()
    x = torch.randn(2, 16).bfloat16()
    for _ in range(2):
        loss = e(x).float().pow(2).mean()
        e.backward(loss)
        e.step()
    e.save_checkpoint(ckpt, tag="t0")
    return _gather_named(e)


def _moe3_load(ckpt, uni):
    import torch
    import torch.distributed as tdist
    from deepspeed_amd.comm import groups
    groups.reset_groups()
    torch.manual_seed(7 + tdist.get_rank())
    e = _moe3_net_engine()
    x = torch.randn(2, 16).bfloat16() + 1.0
    loss = e(x).float().pow(2).mean()
    e.backward(loss)
    e.step()
    e.load_universal_checkpoint(uni)
    return _gather_named(e)


def test_moe_zero3_offline_reassembly_and_universal():
    """Stage-3 MoE: zero_to_fp32 emits every expert under GLOBAL ids
    (per-rank layouts, shard-world-aware concat) and the universal
    round-trip restores rank-correct expert shards."""
    import re
    with tempfile.TemporaryDirectory() as d:
        ckpt, uni = os.path.join(d, "ckpt"), os.path.join(d, "uni")
        saved = run_distributed(_moe3_save, world_size=2, args=(ckpt,))
        from deepspeed_amd.utils.zero_to_fp32 import (
            get_fp32_state_dict_from_zero_checkpoint)
        sd = get_fp32_state_dict_from_zero_checkpoint(ckpt)
        eidx = sorted({int(re.search(r"deepspeed_experts\.(\d+)\.",
                                     k).group(1))
                       for k in sd if "deepspeed_experts" in k})
        assert eidx == [0, 1, 2, 3], eidx
        for r in range(2):  # rank r's local expert i == global r*2+i
            for n, v in saved[r].items():
                m = re.search(r"(.*deepspeed_experts\.)(\d+)(\..*)", n)
                key = (f"{m.group(1)}{r * 2 + int(m.group(2))}{m.group(3)}"
                       if m else n)
                assert torch.allclose(sd[key].float(), v, atol=1e-2), \
                    (r, n, key)
        from deepspeed_amd.checkpoint.universal import ds_to_universal
        names = ds_to_universal(ckpt, uni)
        assert any("@ep" in n for n in names)
        got = run_distributed(_moe3_load, world_size=2, args=(ckpt, uni))
        for r in range(2):
            for n, v in saved[r].items():
                assert torch.allclose(got[r][n], v, atol=1e-2), (r, n)


def test_deepspeed_checkpoint_inspection():
    """DeepSpeedCheckpoint answers topology/content questions offline
    (ref checkpoint/deepspeed_checkpoint.py)."""
    from deepspeed_amd.checkpoint.inspect import DeepSpeedCheckpoint
    with tempfile.TemporaryDirectory() as d:
        run_distributed(_save_and_export, world_size=2, args=(2, d))
        ck = DeepSpeedCheckpoint(d)
        s = ck.summary()
        assert s["zero_stage"] == 2 and s["dp_degree"] == 2
        names = ck.parameter_names()
        assert names and s["n_params"] == len(names)
        got = dict(ck.fp32_parameters())
        assert set(got) == set(names)
        assert "global_steps" in ck.client_state()
