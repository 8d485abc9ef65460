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


def quantize_int4(x, group_size=2048):
    """Symmetric int4 (two nibbles/byte) with per-group scales."""
    from .loader import get_ext, has_ext
    if x.is_cuda and x.dtype == torch.bfloat16 and has_ext():
        return tuple(get_ext().quantize_int4(x.contiguous().view(-1),
                                             group_size)) + (x.shape,)
    flat = x.float().reshape(-1)
    n = flat.numel()
    pad = (-n) % group_size
    g = torch.nn.functional.pad(flat, (0, pad)).reshape(-1, group_size)
    amax = g.abs().amax(dim=1)
    scales = torch.where(amax > 0, amax / 7.0, torch.ones_like(amax))
    qv = torch.clamp(torch.round(g / scales.unsqueeze(1)), -7, 7) \
        .to(torch.int8).reshape(-1)
    lo = qv[0::2] & 0xF
    hi = (qv[1::2] & 0xF) << 4
    packed = (lo | hi).to(torch.uint8)
    return packed, scales, x.shape


def dequantize_int4(q, scales, shape, group_size=2048):
    from .loader import get_ext, has_ext
    numel = 1
    for d in shape:
        numel *= d
    if q.is_cuda and has_ext():
        return get_ext().dequantize_int4(q, scales, group_size,
                                         numel).reshape(shape)
    b = q.to(torch.int16)
    lo = (b & 0xF).to(torch.int8)
    hi = ((b >> 4) & 0xF).to(torch.int8)
    lo = torch.where(lo > 7, lo - 16, lo)
    hi = torch.where(hi > 7, hi - 16, hi)
    v = torch.stack([lo, hi], dim=1).reshape(-1).float()
    npad = v.numel()
    v = v.reshape(-1, group_size) * scales.unsqueeze(1)
    return v.reshape(-1)[:numel].reshape(shape).to(torch.bfloat16)
