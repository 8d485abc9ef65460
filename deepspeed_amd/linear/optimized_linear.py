"""OptimizedLinear: LoRA + (optionally quantized) frozen base weight.

Parity: reference `deepspeed/linear/optimized_linear.py` + `config.py`.
"""
from dataclasses import dataclass

import torch
import torch.nn.functional as F

from .quantization import QuantizedParameter


@dataclass
class LoRAConfig:
    lora_r: int = 64
    lora_alpha: float = 16.0
    base_weight_sharding: int = 1


@dataclass
class QuantizationConfig:
    q_bits: int = 8
    mantissa_bits: int = 3
    group_size: int = 2048


class OptimizedLinear(torch.nn.Module):
    def __init__(self, input_dim, output_dim, lora_config: LoRAConfig = None,
                 quantization_config: QuantizationConfig = None, bias=False,
                 dtype=torch.bfloat16):
        super().__init__()
        self.lora_config = lora_config
        self.quantization_config = quantization_config
        base = torch.nn.Linear(input_dim, output_dim, bias=bias, dtype=dtype)
        if quantization_config is not None:
            quant = "fp8" if quantization_config.mantissa_bits == 3 else "int8"
            self.base_weight = QuantizedParameter(
                base.weight.data, quantization=quant,
                group_size=quantization_config.group_size)
        else:
            self.base_weight = torch.nn.Parameter(base.weight.data,
                                                  requires_grad=False)
        self.bias = base.bias
        if lora_config is not None:
            r = lora_config.lora_r
            self.lora_scaling = lora_config.lora_alpha / r
            self.lora_weight_1 = torch.nn.Linear(input_dim, r, bias=False,
                                                 dtype=dtype)
            self.lora_weight_2 = torch.nn.Linear(r, output_dim, bias=False,
                                                 dtype=dtype)
            torch.nn.init.zeros_(self.lora_weight_2.weight)
        else:
            self.lora_weight_1 = None

    def full_weight(self):
        if isinstance(self.base_weight, QuantizedParameter):
            return self.base_weight.dequantized()
        return self.base_weight

    def forward(self, x):
        w = self.full_weight().to(x.dtype)
        out = F.linear(x, w, self.bias)
        if self.lora_weight_1 is not None:
            out = out + self.lora_scaling * self.lora_weight_2(
                self.lora_weight_1(x))
        return out
