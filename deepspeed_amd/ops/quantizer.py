"""Group-wise quantizer wrappers (int8 / fp8-e4m3) with CPU fallbacks."""
import torch

from .loader import get_ext


def quantize_int8(x, group_size=2048):
    if x.is_cuda:
        q, scales = get_ext(required=True).quantize_int8(
            x.contiguous(), group_size)
        return q, scales
    flat = x.float().reshape(-1, group_size)
    amax = flat.abs().amax(dim=1, keepdim=True)
    scales = torch.where(amax > 0, amax / 127.0, torch.ones_like(amax))
    q = torch.clamp(torch.round(flat / scales), -127, 127).to(torch.int8)
    return q.reshape(-1), scales.reshape(-1)


def dequantize_int8(q, scales, group_size=2048):
    if q.is_cuda:
        return get_ext(required=True).dequantize_int8(q, scales, group_size)
    flat = q.float().reshape(-1, group_size)
    return (flat * scales.reshape(-1, 1)).reshape(-1).to(torch.bfloat16)


def quantize_fp8(x, group_size=2048):
    if x.is_cuda:
        q, scales = get_ext(required=True).quantize_fp8(
            x.contiguous(), group_size)
        return q, scales
    flat = x.float().reshape(-1, group_size)
    amax = flat.abs().amax(dim=1, keepdim=True)
    scales = torch.where(amax > 0, amax / 448.0, torch.ones_like(amax))
    q = (flat / scales).to(torch.float8_e4m3fn).view(torch.uint8)
    return q.reshape(-1), scales.reshape(-1)


def dequantize_fp8(q, scales, group_size=2048):
    if q.is_cuda:
        return get_ext(required=True).dequantize_fp8(q, scales, group_size)
    flat = q.view(torch.float8_e4m3fn).float().reshape(-1, group_size)
    return (flat * scales.reshape(-1, 1)).reshape(-1).to(torch.bfloat16)
