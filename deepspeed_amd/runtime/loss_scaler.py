"""Static and dynamic loss scalers for fp16 training.

Parity: reference `deepspeed/runtime/fp16/loss_scaler.py:131,163,187`.
bf16 training needs no scaler (the engine uses LossScalerBase with scale=1).
"""
import torch

from ..utils.logging import logger


class LossScalerBase:
    def __init__(self, scale=1.0):
        self.cur_scale = scale
        self.dynamic = False

    @property
    def loss_scale(self):
        return self.cur_scale

    def scale_gradient(self, module, grad_in, grad_out):
        return tuple(self.loss_scale * g for g in grad_in)

    def update_scale(self, overflow):
        pass

    def backward(self, loss, retain_graph=False):
        (loss * self.loss_scale).backward(retain_graph=retain_graph)


class LossScaler(LossScalerBase):
    """Static loss scale."""

    def __init__(self, scale=1.0):
        super().__init__(scale)


class DynamicLossScaler(LossScalerBase):
    def __init__(self, init_scale=2**16, scale_factor=2.0, scale_window=1000,
                 min_scale=1.0, delayed_shift=2, consecutive_hysteresis=False,
                 raise_error_at_min_scale=True):
        super().__init__(init_scale)
        self.dynamic = True
        self.cur_iter = 0
        self.last_overflow_iter = -1
        self.scale_factor = scale_factor
        self.scale_window = scale_window
        self.min_scale = min_scale
        self.delayed_shift = delayed_shift
        self.cur_hysteresis = delayed_shift
        self.consecutive_hysteresis = consecutive_hysteresis
        self.raise_error_at_min_scale = raise_error_at_min_scale

    def update_scale(self, overflow):
        if overflow:
            if self.delayed_shift == 1 or self.cur_hysteresis == 1:
                if self.cur_scale == self.min_scale and self.raise_error_at_min_scale:
                    raise Exception(
                        "Current loss scale already at minimum — cannot decrease "
                        "further. Model diverging (inf/nan in gradients).")
                self.cur_scale = max(self.cur_scale / self.scale_factor,
                                     self.min_scale)
            else:
                self.cur_hysteresis -= 1
            self.last_overflow_iter = self.cur_iter
        else:
            if self.consecutive_hysteresis:
                self.cur_hysteresis = self.delayed_shift
            if (self.cur_iter - self.last_overflow_iter) % self.scale_window == 0:
                if not self.consecutive_hysteresis:
                    self.cur_hysteresis = self.delayed_shift
                self.cur_scale *= self.scale_factor
        self.cur_iter += 1


def CreateLossScaler(dtype, static_loss_scale, dynamic_scaling, dynamic_loss_args):
    if dtype == torch.float16 and dynamic_scaling:
        kwargs = dynamic_loss_args or {}
        return DynamicLossScaler(**kwargs)
    if dtype == torch.float16:
        return LossScaler(scale=static_loss_scale)
    return LossScalerBase(1.0)
