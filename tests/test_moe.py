"""MoE: gating invariants, ep=1 correctness, 2-rank a2a dispatch."""
import torch

from tests.common import run_distributed


def test_topk_gating_shapes():
    from deepspeed_amd.moe.sharded_moe import topkgating
    torch.manual_seed(0)
    S, E, k = 64, 8, 2
    logits = torch.randn(S, E)
    l_aux, combine, dispatch, C = topkgating(logits, k, capacity_factor=1.0)
    assert combine.shape == (S, E, C)
    assert dispatch.shape == (S, E, C)
    # each token's combine weights sum to <= 1 (1 unless dropped)
    sums = combine.sum(dim=(1, 2))
    assert (sums <= 1.0 + 1e-5).all()
    assert l_aux.item() > 0
    # capacity respected: each (e, c) slot holds at most one token
    assert (dispatch.float().sum(0) <= 1.0 + 1e-6).all()


def test_moe_single_process_matches_dense_k_all():
    """With num_experts=1, k=1 and huge capacity, MoE == the expert MLP."""
    from deepspeed_amd.moe.layer import MoE
    torch.manual_seed(0)
    M = 16
    expert = torch.nn.Linear(M, M)
    moe = MoE(M, expert, num_experts=1, ep_size=1, k=1, capacity_factor=64)
    x = torch.randn(2, 8, M)
    out, l_aux, _ = moe(x)
    ref = expert(x)
    assert torch.allclose(out, ref, atol=1e-5), (out - ref).abs().max()


def _moe_ep2():
    import torch.distributed as tdist
    from deepspeed_amd.comm import groups
    from deepspeed_amd.moe.layer import MoE
    groups.reset_groups()
    rank = tdist.get_rank()
    torch.manual_seed(7)  # same init on both ranks
    M, E = 16, 4
    expert = torch.nn.Linear(M, M)
    moe = MoE(M, expert, num_experts=E, ep_size=2, k=2, capacity_factor=4.0)
    torch.manual_seed(100 + rank)
    x = torch.randn(2, 8, M, requires_grad=True)
    out, l_aux, _ = moe(x)
    (out.sum() + l_aux).backward()
    assert out.shape == x.shape
    assert x.grad is not None
    # gate weight grads exist (flow through dispatch path)
    assert moe.deepspeed_moe.gate.wg.weight.grad is not None
    return out.sum().item()


def test_moe_ep2_runs():
    results = run_distributed(_moe_ep2, world_size=2)
    assert all(r is not None for r in results)


def _moe_ep_equivalence():
    """ep=2 output must equal ep=1 output with identical experts/tokens."""
    import torch.distributed as tdist
    from deepspeed_amd.comm import groups
    from deepspeed_amd.moe.layer import MoE
    groups.reset_groups()
    rank = tdist.get_rank()
    M, E = 8, 2
    torch.manual_seed(7)
    expert = torch.nn.Linear(M, M)
    moe = MoE(M, expert, num_experts=E, ep_size=2, k=1, capacity_factor=8.0)
    # make local experts differ deterministically: rank r holds expert r
    with torch.no_grad():
        for i, e in enumerate(
                moe.deepspeed_moe.experts.deepspeed_experts):
            e.weight.fill_(0.1 * (rank + 1))
            e.bias.fill_(0.01 * (rank + 1))
    torch.manual_seed(9)  # same input on both ranks
    x = torch.randn(1, 6, M)
    out, _, _ = moe(x)

    # single-process reference: both experts local
    moe_ref = MoE(M, expert, num_experts=E, ep_size=1, k=1,
                  capacity_factor=8.0)
    moe_ref.deepspeed_moe.gate.load_state_dict(
        moe.deepspeed_moe.gate.state_dict())
    with torch.no_grad():
        for i, e in enumerate(
                moe_ref.deepspeed_moe.experts.deepspeed_experts):
            e.weight.fill_(0.1 * (i + 1))
            e.bias.fill_(0.01 * (i + 1))
    ref, _, _ = moe_ref(x)
    assert torch.allclose(out, ref, atol=1e-5), (out - ref).abs().max()
    return True


def test_moe_ep2_equivalence():
    results = run_distributed(_moe_ep_equivalence, world_size=2)
    assert all(results)


def _moe_expert_ckpt():
    import os
    import tempfile
    import torch.distributed as dist
    import deepspeed_amd as ds
    from deepspeed_amd.moe.layer import MoE

    torch.manual_seed(dist.get_rank())
    M = 16

    class Net(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.inp = torch.nn.Linear(M, M)
            self.moe = MoE(M, torch.nn.Linear(M, M), num_experts=4,
                           ep_size=2, k=1, capacity_factor=8.0)
            self.out = torch.nn.Linear(M, 4)

        def forward(self, x):
            h = self.inp(x)
            h, _, _ = self.moe(h)
            return self.out(h)

    cfg = {"train_micro_batch_size_per_gpu": 2,
           "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}},
           "bf16": {"enabled": True},
           "zero_optimization": {"stage": 1}}
    engine, _, _, _ = ds.initialize(model=Net(), config=cfg)
    # make expert weights rank-distinct
    before = {n: p.detach().clone()
              for n, p in engine.module.named_parameters()
              if getattr(p, "group_name", None) is not None}
    assert before, "no expert params found"
    tmp = tempfile.mkdtemp(prefix=f"moeckpt_shared")
    # all ranks must agree on the dir: broadcast rank0's
    obj = [tmp]
    dist.broadcast_object_list(obj, src=0)
    tmp = obj[0]
    engine.save_checkpoint(tmp, tag="t0")
    files = sorted(os.listdir(os.path.join(tmp, "t0")))
    assert any(f.startswith("expert_ep_rank_0") for f in files), files
    assert any(f.startswith("expert_ep_rank_1") for f in files), files
    # perturb experts, then reload and verify restoration
    with torch.no_grad():
        for n, p in engine.module.named_parameters():
            if n in before:
                p.add_(1.0)
    engine.load_checkpoint(tmp, tag="t0")
    for n, p in engine.module.named_parameters():
        if n in before:
            assert torch.allclose(p.detach().float(),
                                  before[n].float(), atol=1e-6), n


def test_moe_expert_checkpoint_ep2():
    run_distributed(_moe_expert_ckpt, world_size=2)


def _moe_zero3_train():
    import torch.distributed as dist
    import deepspeed_amd as ds
    from deepspeed_amd.moe.layer import MoE

    torch.manual_seed(dist.get_rank())
    M = 16

    class Net(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.inp = torch.nn.Linear(M, M)
            self.moe = MoE(M, torch.nn.Linear(M, M), num_experts=4,
                           ep_size=2, k=1, capacity_factor=8.0)
            self.out = torch.nn.Linear(M, 4)

        def forward(self, x):
            h = self.inp(x)
            h, _, _ = self.moe(h)
            return self.out(h)

    cfg = {"train_micro_batch_size_per_gpu": 4,
           "optimizer": {"type": "AdamW", "params": {"lr": 5e-3}},
           "bf16": {"enabled": True},
           "zero_optimization": {"stage": 3, "sub_group_size": 500,
                                 "param_persistence_threshold": 4}}
    engine, _, _, _ = ds.initialize(model=Net(), config=cfg)
    # expert params shard over the (singleton) expert-DP group: full-size
    # local shards, rank-distinct values; dense params shard over DP
    from deepspeed_amd.comm import groups
    epg = groups.get_expert_data_parallel_group("ep_size_2")
    for p in engine.module.parameters():
        if getattr(p, "group_name", None) is not None:
            assert p.ds_group is epg
            assert p.ds_tensor.numel() >= p.ds_numel  # world-1 shard
    torch.manual_seed(7)  # same data on both ranks
    x = torch.randn(4, M).bfloat16()
    y = torch.randn(4, 4).bfloat16()
    losses = []
    for _ in range(6):
        out = engine(x)
        loss = (out - y).float().pow(2).mean()
        engine.backward(loss)
        engine.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0], losses
    # experts remain rank-distinct (no cross-expert mixing)
    w = [p for n, p in engine.module.named_parameters()
         if "deepspeed_experts.0.weight" in n][0]
    from deepspeed_amd.runtime.zero.stage3_params import all_gather_params
    s = w.ds_tensor.float().sum()
    buf = [torch.zeros_like(s) for _ in range(2)]
    dist.all_gather(buf, s)
    assert not torch.allclose(buf[0], buf[1]), "experts were averaged!"
    return losses[-1]


def test_moe_zero3_ep2():
    run_distributed(_moe_zero3_train, world_size=2)


def test_split_params_into_moe_groups():
    from deepspeed_amd.moe.utils import (
        is_moe_param, split_params_into_different_moe_groups_for_optimizer)
    dense = torch.nn.Parameter(torch.randn(4))
    e1 = torch.nn.Parameter(torch.randn(4))
    e1.allreduce = False
    e1.group_name = "ep_size_2"
    e2 = torch.nn.Parameter(torch.randn(4))
    e2.allreduce = False
    e2.group_name = "ep_size_2"
    assert not is_moe_param(dense) and is_moe_param(e1)
    groups = split_params_into_different_moe_groups_for_optimizer(
        {"params": [dense, e1, e2], "lr": 1e-3})
    assert len(groups) == 2
    assert groups[0]["params"] == [dense]
    assert groups[1]["moe"] and groups[1]["name"] == "ep_size_2"
    assert len(groups[1]["params"]) == 2
    # torch optimizer accepts the result
    torch.optim.AdamW(groups)


def _moe_zero3_ckpt():
    import os
    import tempfile
    import torch.distributed as dist
    import deepspeed_amd as ds
    from deepspeed_amd.moe.layer import MoE

    torch.manual_seed(dist.get_rank())
    M = 16

    class Net(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.inp = torch.nn.Linear(M, M)
            self.moe = MoE(M, torch.nn.Linear(M, M), num_experts=4,
                           ep_size=2, k=1, capacity_factor=8.0)
            self.out = torch.nn.Linear(M, 4)

        def forward(self, x):
            h = self.inp(x)
            h, _, _ = self.moe(h)
            return self.out(h)

    cfg = {"train_micro_batch_size_per_gpu": 2,
           "optimizer": {"type": "AdamW", "params": {"lr": 1e-3}},
           "bf16": {"enabled": True},
           "zero_optimization": {"stage": 3, "sub_group_size": 500}}
    engine, _, _, _ = ds.initialize(model=Net(), config=cfg)
    torch.manual_seed(3)
    x = torch.randn(2, M).bfloat16()
    loss = engine(x).float().pow(2).mean()
    engine.backward(loss)
    engine.step()
    # per-rank expert shard values before save
    before = [sg.master32.detach().clone()
              for sg in engine.optimizer.sub_groups]
    tmp = tempfile.mkdtemp()
    obj = [tmp]
    dist.broadcast_object_list(obj, src=0)
    tmp = obj[0]
    engine.save_checkpoint(tmp, tag="t0")
    # ZeRO-3: rank-distinct state flows through zero shards, no expert files
    files = sorted(os.listdir(os.path.join(tmp, "t0")))
    assert not any(f.startswith("expert_ep_rank") for f in files), files
    # another step perturbs; load restores per-rank masters
    loss = engine(x).float().pow(2).mean()
    engine.backward(loss)
    engine.step()
    engine.load_checkpoint(tmp, tag="t0")
    for sg, b in zip(engine.optimizer.sub_groups, before):
        assert torch.allclose(sg.master32.detach(), b, atol=1e-6)


def test_moe_zero3_checkpoint_ep2():
    run_distributed(_moe_zero3_ckpt, world_size=2)


def test_index_dispatch_matches_dense_einsum():
    """The O(S*k) index dispatch/combine must equal the dense [S,E,C]
    one-hot einsum path (same gating decisions, same weighted combine)."""
    import torch
    import torch.nn.functional as F
    from deepspeed_amd.moe.sharded_moe import (topkgating,
                                               topkgating_indices)
    torch.manual_seed(3)
    S, E, M, k = 64, 8, 16, 2
    logits = torch.randn(S, E)
    x = torch.randn(S, M, dtype=torch.float64)

    l1, combine, dispatch, C1 = topkgating(logits, k, capacity_factor=1.0)
    dispatched_d = torch.einsum("sec,sm->ecm", dispatch.double(), x)
    # fake "experts": elementwise transform so routing errors show
    eo_d = dispatched_d * torch.arange(1, E + 1).view(E, 1, 1).double()
    out_d = torch.einsum("sec,ecm->sm", combine.double(), eo_d)

    l2, idx, w, loc, keep, C2 = topkgating_indices(
        logits, k, capacity_factor=1.0)
    assert C1 == C2
    assert torch.allclose(l1, l2)
    flat_pos = idx * C2 + loc
    keep_f = keep.reshape(-1)
    kept_pos = flat_pos.reshape(-1)[keep_f]
    token_idx = torch.arange(S).unsqueeze(1).expand(S, k).reshape(-1)[keep_f]
    dispatched_i = x.new_zeros(E * C2, M).index_copy(
        0, kept_pos, x.index_select(0, token_idx)).reshape(E, C2, M)
    assert torch.allclose(dispatched_i, dispatched_d)
    eo_i = (dispatched_i * torch.arange(1, E + 1).view(E, 1, 1).double()) \
        .reshape(E * C2, M)
    gathered = eo_i.index_select(0, flat_pos.reshape(-1).clamp(max=E*C2-1)) \
        .reshape(S, k, M)
    out_i = ((w * keep).double().unsqueeze(-1) * gathered).sum(1)
    assert torch.allclose(out_i, out_d, atol=1e-12), \
        (out_i - out_d).abs().max()


def test_index_dispatch_large_ec_trains():
    """Shape where the dense [S,E,C] combine tensor would be ~0.5 GB
    (1024 tokens x 64 experts x 2048 cap x fp32) finishes fast and small
    on the index path, and gradients flow to gate + experts."""
    import torch
    from deepspeed_amd.moe.layer import MoE
    torch.manual_seed(0)
    hidden = 32
    moe = MoE(hidden, expert=torch.nn.Linear(hidden, hidden),
              num_experts=64, ep_size=1, k=2, capacity_factor=16.0)
    x = torch.randn(4, 256, hidden, requires_grad=True)
    out, aux, _ = moe(x)
    (out.sum() + 0.01 * aux).backward()
    assert x.grad is not None and x.grad.abs().sum() > 0
    gate_w = moe.deepspeed_moe.gate.wg.weight
    assert gate_w.grad is not None and gate_w.grad.abs().sum() > 0


def _tp_mappings_body():
    """drop_tokens/gather_tokens round-trip across the TP group with
    correct autograd (ref moe/mappings.py parity)."""
    import torch
    import torch.distributed as tdist
    from deepspeed_amd.comm import groups
    from deepspeed_amd.moe.mappings import drop_tokens, gather_tokens
    groups.reset_groups()
    groups.initialize_tensor_parallel(tdist.get_world_size())
    rank = groups.get_tensor_parallel_rank()
    tp = groups.get_tensor_parallel_world_size()
    torch.manual_seed(7)  # same seed everywhere: replicated activations
    x = torch.randn(2, 8, 4, requires_grad=True)
    dropped = drop_tokens(x, dim=1)
    assert dropped.shape[1] == 8 // tp
    assert torch.equal(dropped, x.detach().chunk(tp, dim=1)[rank])
    y = gather_tokens(dropped, dim=1)
    assert torch.allclose(y, x.detach())
    w = torch.randn(2, 8, 4)
    (y * w).sum().backward()
    # gather bwd slices, drop bwd all-gathers: full grad == w
    assert torch.allclose(x.grad, w)
    groups.reset_groups()
    return True


def test_tp_token_mappings_world2():
    from tests.common import run_distributed
    run_distributed(_tp_mappings_body, world_size=2)


def test_tp_token_mappings_identity_no_tp():
    """Without TP configured, drop/gather are identities."""
    from deepspeed_amd.comm import groups
    from deepspeed_amd.moe.mappings import drop_tokens, gather_tokens
    groups.reset_groups()
    import torch
    x = torch.randn(2, 6, 4)
    assert drop_tokens(x) is x and gather_tokens(x) is x


def _moe_expert_global_names():
    """Expert ckpt files carry GLOBAL expert ids: no cross-EP-rank name
    collisions, offline fp32 reassembly keeps every expert, round-trip
    load restores local weights."""
    import os
    import re
    import tempfile
    import torch
    import torch.distributed as dist
    import deepspeed_amd as ds
    from deepspeed_amd.moe.layer import MoE
    torch.manual_seed(dist.get_rank())
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
           "zero_optimization": {"stage": 2}}
    engine, _, _, _ = ds.initialize(model=Net(), config=cfg)
    x = torch.randn(2, M).bfloat16()
    loss = engine(x).float().pow(2).mean()
    engine.backward(loss)
    engine.step()
    want = {n: p.detach().float().clone()
            for n, p in engine.module.named_parameters()}
    tmp = tempfile.mkdtemp()
    obj = [tmp]
    dist.broadcast_object_list(obj, src=0)
    tmp = obj[0]
    engine.save_checkpoint(tmp, tag="t0")
    if dist.get_rank() == 0:
        e0 = torch.load(os.path.join(
            tmp, "t0", "expert_ep_rank_0_mp_rank_00_model_states.pt"),
            weights_only=False)
        e1 = torch.load(os.path.join(
            tmp, "t0", "expert_ep_rank_1_mp_rank_00_model_states.pt"),
            weights_only=False)
        k0, k1 = set(e0["module"]), set(e1["module"])
        assert k0.isdisjoint(k1), ("collision", k0 & k1)
        idxs = sorted({int(re.search(r"deepspeed_experts\.(\d+)\.",
                                     k).group(1)) for k in (k0 | k1)})
        assert idxs == [0, 1, 2, 3], idxs
        from deepspeed_amd.utils.zero_to_fp32 import (
            get_fp32_state_dict_from_zero_checkpoint)
        sd = get_fp32_state_dict_from_zero_checkpoint(tmp)
        got = {int(re.search(r"deepspeed_experts\.(\d+)\.", k).group(1))
               for k in sd if "deepspeed_experts" in k}
        assert got == {0, 1, 2, 3}, got
    loss = engine(x).float().pow(2).mean()
    engine.backward(loss)
    engine.step()
    engine.load_checkpoint(tmp, tag="t0")
    for n, p in engine.module.named_parameters():
        assert torch.allclose(p.detach().float(), want[n], atol=1e-2), n
    return True


def test_moe_expert_global_names_world2():
    from tests.common import run_distributed
    assert all(run_distributed(_moe_expert_global_names, world_size=2))
