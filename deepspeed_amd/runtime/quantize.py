"""MoQ: Mixture-of-Quantization training quantizer.

Parity: reference `runtime/quantize.py:14` (Quantizer). Weights fake-
quantize IN TRAINING with a per-layer bit width that anneals from
start_bits to target_bits over a period; with eigenvalue modulation,
layers with larger Hessian eigenvalues (more sensitive) anneal SLOWER
(their period stretches by the normalized eigenvalue), which is the MoQ
paper's scheduling rule.
"""
import torch

from ..compression import fake_quantize
from ..utils.logging import log_dist


class Quantizer:
    def __init__(self, q_groups=1, q_mixed_fp16=False, q_change_ratio=0.01,
                 q_type=0, q_rounding=0, q_verbose=False,
                 q_eigenvalue=False, use_quantizer_kernel=False,
                 layer_num=0, start_bits=16, target_bits=8,
                 quantize_period=1000):
        self.q_groups = q_groups
        self.q_mixed_fp16 = q_mixed_fp16
        self.q_change_ratio = q_change_ratio
        self.q_verbose = q_verbose
        self.q_eigenvalue = q_eigenvalue
        self.layer_num = max(layer_num, 1)
        self.start_bits = start_bits
        self.target_bits = target_bits
        self.base_period = quantize_period
        self.qsteps = 0
        self.bits = [start_bits] * self.layer_num
        self.periods = [quantize_period] * self.layer_num

    def step(self):
        self.qsteps += 1

    def any_precision_switch(self):
        return any(b != self.target_bits for b in self.bits)

    def _update_bits(self, block_eigenvalue=None):
        """Halve the bit width each elapsed period; sensitive layers
        (higher eigenvalue) stretch their period."""
        if block_eigenvalue:
            vals = [abs(block_eigenvalue.get(i, (0.0, 0))[0]
                        if isinstance(block_eigenvalue.get(i, 0.0), tuple)
                        else block_eigenvalue.get(i, 0.0))
                    for i in range(self.layer_num)]
            mx = max(vals) or 1.0
            self.periods = [int(self.base_period * (1.0 + v / mx))
                            for v in vals]
        for i in range(self.layer_num):
            if self.bits[i] > self.target_bits and \
                    self.qsteps >= self.periods[i] * \
                    (1 + self._halvings(i)):
                self.bits[i] = max(self.target_bits, self.bits[i] // 2)
                if self.q_verbose:
                    log_dist(f"MoQ: layer {i} -> {self.bits[i]} bits at "
                             f"step {self.qsteps}", ranks=[0])

    def _halvings(self, i):
        b, n = self.start_bits, 0
        while b > self.bits[i]:
            b //= 2
            n += 1
        return n

    @torch.no_grad()
    def quantize(self, parameter_group, overflow=False,
                 eigenvalue_enabled=False, block_eigenvalue=None):
        """Fake-quantize each layer's parameters at its current bit
        width (skipped on fp16 overflow steps, ref quantize:52)."""
        if overflow and not eigenvalue_enabled:
            return
        self.step()
        self._update_bits(block_eigenvalue if eigenvalue_enabled else None)
        for i, params in enumerate(parameter_group):
            bits = self.bits[i % self.layer_num]
            if bits >= 16:
                continue
            plist = params if isinstance(params, (list, tuple)) else [params]
            for p in plist:
                p.data.copy_(fake_quantize(p.data.float(), bits)
                             .to(p.dtype))
