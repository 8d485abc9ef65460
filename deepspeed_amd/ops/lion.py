"""Fused Lion optimizers for MI355X (GPU) and the EPYC host (offload).

Parity: reference `deepspeed/ops/lion/fused_lion.py` (FusedLion,
csrc/lion/multi_tensor_lion.cu) and `cpu_lion.py` (DeepSpeedCPULion,
csrc/lion/cpu_lion_impl.cpp). The GPU path is one grid-stride HIP kernel
per fp32 master slab (sign update is pure-bandwidth work); the host path
is the OpenMP `cpu_lion_step`.
"""
import torch

from .loader import has_ext, get_ext


def _lion_torch(p, g, m, lr, beta1, beta2, wd):
    p.mul_(1 - lr * wd)
    u = m.mul(beta1).add_(g, alpha=1 - beta1)
    p.add_(torch.sign(u), alpha=-lr)
    m.mul_(beta2).add_(g, alpha=1 - beta2)


class FusedLion(torch.optim.Optimizer):
    """GPU Lion on fp32 master weights (HIP lion_kernel_f32)."""

    def __init__(self, params, lr=1e-4, betas=(0.9, 0.99), weight_decay=0.0):
        defaults = dict(lr=lr, betas=betas, weight_decay=weight_decay)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            beta1, beta2 = group["betas"]
            ps, gs, ms = [], [], []
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["exp_avg"] = torch.zeros_like(
                        p, dtype=torch.float32)
                ps.append(p)
                gs.append(p.grad)
                ms.append(state["exp_avg"])
            if not ps:
                continue
            if ps[0].is_cuda and ps[0].dtype == torch.float32 and has_ext():
                get_ext().multi_tensor_lion(ps, gs, ms, group["lr"], beta1,
                                            beta2, group["weight_decay"], [])
            else:
                for p, g, m in zip(ps, gs, ms):
                    _lion_torch(p.data, g.float(), m, group["lr"], beta1,
                                beta2, group["weight_decay"])
        return loss


class DeepSpeedCPULion(torch.optim.Optimizer):
    """Host-side Lion for ZeRO-Offload (OpenMP cpu_lion_step)."""

    def __init__(self, model_params, lr=1e-4, betas=(0.9, 0.99),
                 weight_decay=0.0, fp32_optimizer_states=True):
        defaults = dict(lr=lr, betas=betas, weight_decay=weight_decay)
        super().__init__(model_params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        use_ext = has_ext()
        for group in self.param_groups:
            beta1, beta2 = group["betas"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                assert not p.is_cuda, "CPULion expects host tensors"
                state = self.state[p]
                if len(state) == 0:
                    state["exp_avg"] = torch.zeros_like(
                        p, dtype=torch.float32)
                if use_ext and p.dtype == torch.float32:
                    get_ext().cpu_lion_step(
                        p.data, p.grad, state["exp_avg"], None, group["lr"],
                        beta1, beta2, group["weight_decay"])
                else:
                    _lion_torch(p.data, p.grad.float(), state["exp_avg"],
                                group["lr"], beta1, beta2,
                                group["weight_decay"])
        return loss
