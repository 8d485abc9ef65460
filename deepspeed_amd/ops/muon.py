"""Muon optimizer: orthogonalized momentum for 2-D weights.

Parity: reference `runtime/zero/muon/` (original_muon.py + stage3
integration). Newton-Schulz iteration orthogonalizes the momentum update;
non-2D params fall back to AdamW.
"""
import torch


@torch.no_grad()
def zeropower_via_newtonschulz5(G, steps=5, eps=1e-7):
    """Approximate UV^T of the SVD of G (quintic Newton-Schulz)."""
    a, b, c = (3.4445, -4.7750, 2.0315)
    X = G.float()
    transposed = False
    if X.shape[0] > X.shape[1]:
        X = X.T
        transposed = True
    X = X / (X.norm() + eps)
    for _ in range(steps):
        A = X @ X.T
        B = b * A + c * A @ A
        X = a * X + B @ X
    if transposed:
        X = X.T
    return X


class Muon(torch.optim.Optimizer):
    def __init__(self, params, lr=2e-2, momentum=0.95, nesterov=True,
                 ns_steps=5, adamw_lr=3e-4, adamw_betas=(0.9, 0.95),
                 adamw_eps=1e-8, weight_decay=0.0):
        defaults = dict(lr=lr, momentum=momentum, nesterov=nesterov,
                        ns_steps=ns_steps, adamw_lr=adamw_lr,
                        adamw_betas=adamw_betas, adamw_eps=adamw_eps,
                        weight_decay=weight_decay)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            group["step"] = group.get("step", 0) + 1
            for p in group["params"]:
                if p.grad is None:
                    continue
                g = p.grad.float()
                state = self.state[p]
                use_muon = p.dim() == 2 and min(p.shape) > 1
                if use_muon:
                    if "momentum_buffer" not in state:
                        state["momentum_buffer"] = torch.zeros_like(g)
                    buf = state["momentum_buffer"]
                    buf.mul_(group["momentum"]).add_(g)
                    eff = (g.add(buf, alpha=group["momentum"])
                           if group["nesterov"] else buf)
                    u = zeropower_via_newtonschulz5(eff, group["ns_steps"])
                    scale = max(1.0, p.shape[0] / p.shape[1]) ** 0.5
                    if group["weight_decay"]:
                        p.data.mul_(1 - group["lr"] * group["weight_decay"])
                    p.data.add_(u.to(p.dtype), alpha=-group["lr"] * scale)
                else:
                    # AdamW fallback for vectors/scalars
                    if "exp_avg" not in state:
                        state["exp_avg"] = torch.zeros_like(g)
                        state["exp_avg_sq"] = torch.zeros_like(g)
                    b1, b2 = group["adamw_betas"]
                    t = group["step"]
                    m, v = state["exp_avg"], state["exp_avg_sq"]
                    m.mul_(b1).add_(g, alpha=1 - b1)
                    v.mul_(b2).addcmul_(g, g, value=1 - b2)
                    denom = (v / (1 - b2**t)).sqrt_().add_(group["adamw_eps"])
                    p.data.addcdiv_(m.to(p.dtype), denom.to(p.dtype),
                                    value=-group["adamw_lr"] / (1 - b1**t))
        return loss
