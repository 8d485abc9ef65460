"""DeepSpeedCPUAdam — host-side AdamW for ZeRO-Offload.

Parity: reference `deepspeed/ops/adam/cpu_adam.py:13`. Uses the OpenMP
cpu_adam_step from the in-tree extension when built, else a torch fallback.
"""
import torch

from .loader import has_ext, get_ext


class DeepSpeedCPUAdam(torch.optim.Optimizer):
    def __init__(self, model_params, lr=1e-3, bias_correction=True,
                 betas=(0.9, 0.999), eps=1e-8, weight_decay=0.0,
                 amsgrad=False, adamw_mode=True, fp32_optimizer_states=True):
        if amsgrad:
            raise RuntimeError("CPUAdam does not support amsgrad")
        defaults = dict(lr=lr, bias_correction=bias_correction, betas=betas,
                        eps=eps, weight_decay=weight_decay)
        super().__init__(model_params, defaults)
        self.adam_w_mode = 1 if adamw_mode else 0
        self._grad_scale = 1.0
        self._out16 = {}

    def set_grad_scale(self, scale):
        self._grad_scale = float(scale)

    def set_fused_out16(self, mapping):
        self._out16 = dict(mapping)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        use_ext = has_ext()
        for group in self.param_groups:
            if "step" not in group:
                group["step"] = 0
            group["step"] += 1
            beta1, beta2 = group["betas"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                assert not p.is_cuda, "CPUAdam expects host tensors"
                state = self.state[p]
                if len(state) == 0:
                    state["exp_avg"] = torch.zeros_like(
                        p, dtype=torch.float32)
                    state["exp_avg_sq"] = torch.zeros_like(
                        p, dtype=torch.float32)
                out16 = self._out16.get(p)
                if use_ext:
                    get_ext().cpu_adam_step(
                        p.data.view(-1), p.grad.view(-1),
                        state["exp_avg"].view(-1),
                        state["exp_avg_sq"].view(-1),
                        out16.view(-1) if out16 is not None else None,
                        group["lr"], beta1, beta2, group["eps"],
                        group["step"], self.adam_w_mode,
                        1 if group.get("bias_correction", True) else 0,
                        group["weight_decay"], self._grad_scale)
                else:
                    self._torch_step(p, state, group, beta1, beta2, out16)
        return loss

    def _torch_step(self, p, state, group, beta1, beta2, out16):
        step = group["step"]
        bc1 = 1 - beta1**step if group.get("bias_correction", True) else 1.0
        bc2 = 1 - beta2**step if group.get("bias_correction", True) else 1.0
        g = p.grad.float() * self._grad_scale
        if group["weight_decay"] != 0.0 and self.adam_w_mode:
            p.data.mul_(1.0 - group["lr"] * group["weight_decay"])
        elif group["weight_decay"] != 0.0:
            g = g.add(p.data, alpha=group["weight_decay"])
        m, v = state["exp_avg"], state["exp_avg_sq"]
        m.mul_(beta1).add_(g, alpha=1 - beta1)
        v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
        denom = (v / bc2).sqrt_().add_(group["eps"])
        p.data.addcdiv_(m, denom, value=-group["lr"] / bc1)
        if out16 is not None:
            out16.copy_(p.data.view(-1).to(out16.dtype))
