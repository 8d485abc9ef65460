"""FusedAdam — multi-tensor AdamW on MI355X.

Parity: reference `deepspeed/ops/adam/fused_adam.py:18` +
`csrc/adam/multi_tensor_adam.cu:129`. MI355X-native: the HIP kernel
(`csrc/multi_tensor_adam.hip`) is bandwidth-bound, uses dwordx4 vector IO
and a grid-stride chunk loop sized for 256 CUs; tensor metadata is passed
via device-side pointer arrays (no 4KB kernarg struct repacking per launch).

CPU fallback (tests / no-GPU dev): torch._foreach fused math, same update
rule, fp32 state.
"""
import torch

from .loader import get_ext


class FusedAdam(torch.optim.Optimizer):
    def __init__(self, params, lr=1e-3, bias_correction=True,
                 betas=(0.9, 0.999), eps=1e-8, adam_w_mode=True,
                 weight_decay=0.0, amsgrad=False, set_grad_none=True):
        if amsgrad:
            raise RuntimeError("FusedAdam does not support amsgrad")
        defaults = dict(lr=lr, bias_correction=bias_correction, betas=betas,
                        eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)
        self.adam_w_mode = 1 if adam_w_mode else 0
        self.set_grad_none = set_grad_none
        # ZeRO fusion hooks: grad scale folded into the kernel (no extra
        # pass over grads) and bf16 shard writeback fused into the update.
        self._grad_scale = 1.0
        self._out16 = {}  # param -> flat bf16 tensor (same numel)

    def set_grad_scale(self, scale):
        self._grad_scale = float(scale)

    def set_fused_out16(self, mapping):
        self._out16 = dict(mapping)

    def zero_grad(self, set_to_none=True):
        if self.set_grad_none or set_to_none:
            for group in self.param_groups:
                for p in group["params"]:
                    p.grad = None
        else:
            super().zero_grad(set_to_none=False)

    @torch.no_grad()
    def step(self, closure=None, grads=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        for group in self.param_groups:
            params, grads_, exp_avgs, exp_avg_sqs = [], [], [], []
            if "step" not in group:
                group["step"] = 0
            group["step"] += 1
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["exp_avg"] = torch.zeros_like(p, dtype=torch.float32)
                    state["exp_avg_sq"] = torch.zeros_like(p, dtype=torch.float32)
                params.append(p)
                grads_.append(p.grad)
                exp_avgs.append(state["exp_avg"])
                exp_avg_sqs.append(state["exp_avg_sq"])
            if not params:
                continue

            beta1, beta2 = group["betas"]
            step = group["step"]
            bias_correction = 1 if group.get("bias_correction", True) else 0

            if params[0].is_cuda:
                ext = get_ext(required=True)
                out16 = []
                if self._out16 and all(p in self._out16 for p in params):
                    out16 = [self._out16[p] for p in params]
                ext.multi_tensor_adam(
                    params, grads_, exp_avgs, exp_avg_sqs, group["lr"], beta1,
                    beta2, group["eps"], step, self.adam_w_mode,
                    bias_correction, group["weight_decay"], out16,
                    self._grad_scale)
            else:
                if self._grad_scale != 1.0:
                    for g in grads_:
                        g.mul_(self._grad_scale)
                self._cpu_step(params, grads_, exp_avgs, exp_avg_sqs,
                               group["lr"], beta1, beta2, group["eps"], step,
                               bias_correction, group["weight_decay"])
                for p in params:
                    if p in self._out16:
                        self._out16[p].copy_(p.detach().reshape(-1))
        return loss

    def _cpu_step(self, params, grads, exp_avgs, exp_avg_sqs, lr, beta1,
                  beta2, eps, step, bias_correction, weight_decay):
        bc1 = 1 - beta1**step if bias_correction else 1.0
        bc2 = 1 - beta2**step if bias_correction else 1.0
        for p, g, m, v in zip(params, grads, exp_avgs, exp_avg_sqs):
            g32 = g.float()
            p32 = p.float()
            if weight_decay != 0.0 and self.adam_w_mode:
                p32.mul_(1.0 - lr * weight_decay)
            elif weight_decay != 0.0:
                g32 = g32.add(p32, alpha=weight_decay)
            m.mul_(beta1).add_(g32, alpha=1 - beta1)
            v.mul_(beta2).addcmul_(g32, g32, value=1 - beta2)
            denom = (v / bc2).sqrt_().add_(eps)
            p32.addcdiv_(m, denom, value=-lr / bc1)
            p.copy_(p32)


# DeepSpeed-compat alias
DeepSpeedCPUAdam = None  # defined in cpu_adam.py
