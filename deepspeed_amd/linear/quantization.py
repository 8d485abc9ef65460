"""Quantized weight storage: fp8/int8 group-wise parameters.

Parity: reference `deepspeed/linear/quantization.py` (QuantizedParameter).
Weights live quantized (1 byte/elem + fp32 group scales) and dequantize
on the fly at forward.
"""
import torch

from ..ops import quantizer as Q


class QuantizedParameter(torch.nn.Parameter):
    """Holds quantized bytes; .dequantized() returns bf16."""

    def __new__(cls, data, requires_grad=False, quantization="fp8",
                group_size=2048):
        numel = data.numel()
        if numel % group_size != 0:
            group_size = 1
            while numel % (group_size * 2) == 0 and group_size < 2048:
                group_size *= 2
        if quantization == "fp8":
            qdata, scales = Q.quantize_fp8(
                data.detach().to(torch.bfloat16).reshape(-1), group_size)
        else:
            qdata, scales = Q.quantize_int8(
                data.detach().to(torch.bfloat16).reshape(-1), group_size)
        self = super().__new__(cls, qdata, requires_grad=False)
        self.quantization = quantization
        self.group_size = group_size
        self.scales = scales
        self.orig_shape = tuple(data.shape)
        return self

    def dequantized(self):
        if self.quantization == "fp8":
            out = Q.dequantize_fp8(self.data, self.scales, self.group_size)
        else:
            out = Q.dequantize_int8(self.data, self.scales, self.group_size)
        return out.reshape(self.orig_shape)


class QuantizedLinear(torch.nn.Module):
    def __init__(self, linear, quantization="fp8", group_size=2048):
        super().__init__()
        self.weight = QuantizedParameter(linear.weight.data,
                                         quantization=quantization,
                                         group_size=group_size)
        self.bias = linear.bias
        self.out_features = linear.out_features
        self.in_features = linear.in_features

    def forward(self, x):
        w = self.weight.dequantized().to(x.dtype)
        b = self.bias.to(x.dtype) if self.bias is not None else None
        return torch.nn.functional.linear(x, w, b)
