"""Adagrad optimizers: MI355X fused kernel + host (offload) OpenMP step.

Parity: reference `deepspeed/ops/adagrad/cpu_adagrad.py`
(DeepSpeedCPUAdagrad, csrc/adagrad/cpu_adagrad.cpp). The GPU variant is
the same grid-stride slab pattern as FusedAdam.
"""
import torch

from .loader import has_ext, get_ext


def _adagrad_torch(p, g, h, lr, eps, wd):
    if wd:
        g = g.add(p, alpha=wd)
    h.addcmul_(g, g)
    p.addcdiv_(g, h.sqrt().add_(eps), value=-lr)


class FusedAdagrad(torch.optim.Optimizer):
    """GPU Adagrad on fp32 masters (HIP adagrad_kernel_f32)."""

    def __init__(self, params, lr=1e-2, eps=1e-8, weight_decay=0.0):
        defaults = dict(lr=lr, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            ps, gs, hs = [], [], []
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["sq_accum"] = torch.zeros_like(
                        p, dtype=torch.float32)
                ps.append(p)
                gs.append(p.grad)
                hs.append(state["sq_accum"])
            if not ps:
                continue
            if ps[0].is_cuda and ps[0].dtype == torch.float32 and has_ext():
                get_ext().multi_tensor_adagrad(
                    ps, gs, hs, group["lr"], group["eps"],
                    group["weight_decay"], [])
            else:
                for p, g, h in zip(ps, gs, hs):
                    _adagrad_torch(p.data, g.float(), h, group["lr"],
                                   group["eps"], group["weight_decay"])
        return loss


class DeepSpeedCPUAdagrad(torch.optim.Optimizer):
    """Host-side Adagrad for ZeRO-Offload (OpenMP cpu_adagrad_step)."""

    def __init__(self, model_params, lr=1e-2, eps=1e-8, weight_decay=0.0,
                 fp32_optimizer_states=True):
        defaults = dict(lr=lr, eps=eps, weight_decay=weight_decay)
        super().__init__(model_params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        use_ext = has_ext()
        for group in self.param_groups:
            for p in group["params"]:
                if p.grad is None:
                    continue
                assert not p.is_cuda, "CPUAdagrad expects host tensors"
                state = self.state[p]
                if len(state) == 0:
                    state["sq_accum"] = torch.zeros_like(
                        p, dtype=torch.float32)
                if use_ext and p.dtype == torch.float32:
                    get_ext().cpu_adagrad_step(
                        p.data, p.grad, state["sq_accum"], None, group["lr"],
                        group["eps"], group["weight_decay"])
                else:
                    _adagrad_torch(p.data, p.grad.float(), state["sq_accum"],
                                   group["lr"], group["eps"],
                                   group["weight_decay"])
        return loss
