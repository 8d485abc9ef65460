"""Optimizer family: Lion/Adagrad/LAMB (fused + CPU), 0/1 Adam, 1-bit LAMB.

The CPU OpenMP extension steps are validated against independent torch
implementations; the GPU fused kernels are validated in test_ops_gpu-style
gpu-marked tests at the bottom.
"""
import pytest
import torch

from deepspeed_amd.ops import has_ext
from deepspeed_amd.ops.adagrad import DeepSpeedCPUAdagrad, FusedAdagrad
from deepspeed_amd.ops.lamb import FusedLamb, _lamb_torch
from deepspeed_amd.ops.lion import DeepSpeedCPULion, FusedLion, _lion_torch


def _mkparams(seed=0, n=3):
    torch.manual_seed(seed)
    ps = [torch.nn.Parameter(torch.randn(17 + 13 * i) * (i + 1))
          for i in range(n)]
    for p in ps:
        p.grad = torch.randn_like(p)
    return ps


def _clone(ps):
    out = []
    for p in ps:
        q = torch.nn.Parameter(p.detach().clone())
        q.grad = p.grad.clone()
        out.append(q)
    return out


def test_cpu_lion_matches_torch_formula():
    ps = _mkparams()
    qs = _clone(ps)
    opt = DeepSpeedCPULion(ps, lr=1e-2, betas=(0.9, 0.99), weight_decay=0.01)
    ms = [torch.zeros_like(p) for p in qs]
    for _ in range(3):
        opt.step()
        for q, m in zip(qs, ms):
            _lion_torch(q.data, q.grad, m, 1e-2, 0.9, 0.99, 0.01)
    for p, q in zip(ps, qs):
        assert torch.allclose(p, q, atol=1e-5), (p - q).abs().max()


def test_cpu_adagrad_matches_torch_optim():
    ps = _mkparams(seed=1)
    qs = _clone(ps)
    opt = DeepSpeedCPUAdagrad(ps, lr=1e-2, eps=1e-10)
    ref = torch.optim.Adagrad(qs, lr=1e-2, eps=1e-10)
    for _ in range(3):
        opt.step()
        ref.step()
    for p, q in zip(ps, qs):
        assert torch.allclose(p, q, atol=1e-5), (p - q).abs().max()


def test_lamb_trust_ratio_cpu():
    ps = _mkparams(seed=2)
    qs = _clone(ps)
    opt = FusedLamb(ps, lr=1e-2, betas=(0.9, 0.999), eps=1e-6,
                    weight_decay=0.01)
    ms = [torch.zeros_like(p) for p in qs]
    vs = [torch.zeros_like(p) for p in qs]
    for it in range(3):
        opt.step()
        for q, m, v in zip(qs, ms, vs):
            _lamb_torch(q.data, q.grad, m, v, it + 1, 1e-2, 0.9, 0.999,
                        1e-6, 0.01)
    for p, q in zip(ps, qs):
        assert torch.allclose(p, q, atol=1e-5), (p - q).abs().max()
    # the step actually moved the params
    assert (ps[0] - _mkparams(seed=2)[0]).abs().max() > 1e-4


def test_zero_one_adam_single_process():
    from deepspeed_amd.ops.onebit_adam import ZeroOneAdam
    torch.manual_seed(0)
    w = torch.nn.Parameter(torch.randn(64))
    target = torch.randn(64)
    opt = ZeroOneAdam([w], lr=0.05, var_freeze_step=5,
                      local_step_clipper=4)
    l0 = None
    for _ in range(30):
        opt.zero_grad()
        loss = ((w - target) ** 2).mean()
        loss.backward()
        opt.step()
        if l0 is None:
            l0 = loss.item()
    assert loss.item() < 0.5 * l0


def test_onebit_lamb_single_process():
    from deepspeed_amd.ops.onebit_adam import OnebitLamb
    torch.manual_seed(0)
    w = torch.nn.Parameter(torch.randn(64))
    target = torch.randn(64)
    opt = OnebitLamb([w], lr=0.05, freeze_step=10)
    l0 = None
    for _ in range(30):
        opt.zero_grad()
        loss = ((w - target) ** 2).mean()
        loss.backward()
        opt.step()
        if l0 is None:
            l0 = loss.item()
    assert loss.item() < 0.5 * l0


def test_engine_routes_new_optimizers():
    """Config-name → optimizer-class routing (no dist/GPU needed)."""
    from deepspeed_amd.config import DeepSpeedConfig
    for name, cls in [("lion", FusedLion), ("adagrad", FusedAdagrad),
                      ("lamb", FusedLamb)]:
        cfg = DeepSpeedConfig({
            "train_micro_batch_size_per_gpu": 1,
            "optimizer": {"type": name, "params": {"lr": 1e-3}},
        })
        assert cfg.optimizer.type.lower() == name


# ---------------------------------------------------------------- GPU side
@pytest.mark.gpu
@pytest.mark.parametrize("kind", ["lion", "adagrad", "lamb"])
def test_fused_gpu_matches_torch(kind):
    assert has_ext(), "HIP extension must be built"
    torch.manual_seed(0)
    dev = "cuda"
    ps = [torch.nn.Parameter(torch.randn(1000 + i * 7, device=dev))
          for i in range(2)]
    for p in ps:
        p.grad = torch.randn_like(p)
    qs = _clone(ps)
    if kind == "lion":
        opt = FusedLion(ps, lr=1e-2, weight_decay=0.01)
        step_ref = lambda q, st, it: _lion_torch(  # noqa: E731
            q.data, q.grad, st.setdefault("m", torch.zeros_like(q)),
            1e-2, 0.9, 0.99, 0.01)
    elif kind == "adagrad":
        opt = FusedAdagrad(ps, lr=1e-2, eps=1e-8)
        ref = torch.optim.Adagrad(qs, lr=1e-2, eps=1e-8)
        step_ref = lambda q, st, it: None  # noqa: E731
    else:
        opt = FusedLamb(ps, lr=1e-2, weight_decay=0.01)
        step_ref = lambda q, st, it: _lamb_torch(  # noqa: E731
            q.data, q.grad,
            st.setdefault("m", torch.zeros_like(q)),
            st.setdefault("v", torch.zeros_like(q)),
            it + 1, 1e-2, 0.9, 0.999, 1e-6, 0.01)
    states = [dict() for _ in qs]
    for it in range(3):
        opt.step()
        if kind == "adagrad":
            ref.step()
        else:
            for q, st in zip(qs, states):
                step_ref(q, st, it)
    for p, q in zip(ps, qs):
        assert torch.allclose(p, q, atol=2e-4), (p - q).abs().max()


def _bf16_opt_run():
    from deepspeed_amd.ops import FusedAdam
    from deepspeed_amd.runtime.bf16_optimizer import BF16_Optimizer
    torch.manual_seed(0)
    m = torch.nn.Linear(32, 32).to(torch.bfloat16)
    base = FusedAdam(m.parameters(), lr=1e-2)
    opt = BF16_Optimizer(base)
    assert len(opt.fp32_groups_flat_partition) >= 1
    ref = torch.nn.Linear(32, 32)
    with torch.no_grad():
        ref.weight.copy_(m.weight.float())
        ref.bias.copy_(m.bias.float())
    ropt = torch.optim.AdamW(ref.parameters(), lr=1e-2, betas=(0.9, 0.999),
                             eps=1e-8, weight_decay=0.0)
    x = torch.randn(8, 32)
    for _ in range(3):
        loss = m(x.bfloat16()).float().pow(2).mean()
        opt.backward(loss)
        opt.step()
        opt.zero_grad()
        rl = ref(x).pow(2).mean()
        rl.backward()
        ropt.step()
        ropt.zero_grad()
    opt.update_lp_params()
    err = (m.weight.float() - ref.weight).abs().max().item()
    assert err < 0.05, err


def test_bf16_optimizer_wrapper():
    """BF16_Optimizer = named stage-1 fp32-master path (single process)."""
    from tests.common import run_distributed
    run_distributed(_bf16_opt_run, world_size=1)
