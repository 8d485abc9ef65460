"""Fused LAMB for MI355X.

Parity: reference `deepspeed/ops/lamb/fused_lamb.py`
(csrc/lamb/fused_lamb_cuda_kernel.cu). MI355X-native design: the trust
ratio (||w|| / ||update||) is computed DEVICE-side — phase 1 updates the
Adam moments, writes the update direction into the grad buffer and
atomically accumulates the two squared norms into a workspace; phase 2
applies p -= lr * ratio * u reading the workspace. No host sync per
parameter, so the step stays one stream of back-to-back kernels.
"""
import torch

from .loader import has_ext, get_ext


def _lamb_torch(p, g, m, v, step, lr, beta1, beta2, eps, wd):
    m.mul_(beta1).add_(g, alpha=1 - beta1)
    v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
    bc1 = 1 - beta1 ** step
    bc2 = 1 - beta2 ** step
    u = (m / bc1) / ((v / bc2).sqrt() + eps) + wd * p
    r1 = p.norm()
    r2 = u.norm()
    ratio = (r1 / r2).item() if (r1 > 0 and r2 > 0) else 1.0
    p.add_(u, alpha=-lr * ratio)


class FusedLamb(torch.optim.Optimizer):
    def __init__(self, params, lr=1e-3, bias_correction=True,
                 betas=(0.9, 0.999), eps=1e-6, weight_decay=0.0):
        defaults = dict(lr=lr, bias_correction=bias_correction, betas=betas,
                        eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)
        self._ws = None

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            if "step" not in group:
                group["step"] = 0
            group["step"] += 1
            beta1, beta2 = group["betas"]
            ps, gs, us, ms, vs = [], [], [], [], []
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["exp_avg"] = torch.zeros_like(
                        p, dtype=torch.float32)
                    state["exp_avg_sq"] = torch.zeros_like(
                        p, dtype=torch.float32)
                    if p.is_cuda:
                        state["update"] = torch.empty_like(
                            p, dtype=torch.float32)
                ps.append(p)
                gs.append(p.grad)
                us.append(state.get("update"))
                ms.append(state["exp_avg"])
                vs.append(state["exp_avg_sq"])
            if not ps:
                continue
            if ps[0].is_cuda and ps[0].dtype == torch.float32 and has_ext():
                if self._ws is None or self._ws.numel() < 2 * len(ps):
                    self._ws = torch.zeros(2 * len(ps), dtype=torch.float32,
                                           device=ps[0].device)
                else:
                    self._ws.zero_()
                get_ext().multi_tensor_lamb(
                    ps, gs, us, ms, vs, self._ws, group["lr"], beta1,
                    beta2, group["eps"], group["step"],
                    1 if group["bias_correction"] else 0,
                    group["weight_decay"], [], 1.0)
            else:
                for p, g, m, v in zip(ps, gs, ms, vs):
                    _lamb_torch(p.data, g.float(), m, v, group["step"],
                                group["lr"], beta1, beta2, group["eps"],
                                group["weight_decay"])
        return loss
