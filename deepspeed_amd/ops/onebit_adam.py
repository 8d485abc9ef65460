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
