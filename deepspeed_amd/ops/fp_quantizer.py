"""FP_Quantize — group-wise low-precision float quantization for MI355X.

Parity: reference `deepspeed/ops/fp_quantizer/quantize.py` (FP_Quantize,
csrc/fp_quantizer; q_bits 4/6/8/12, used by MoE inference and comms
compression). Formats follow OCP MX element conventions (saturating, no
inf/nan): fp4 = e2m1 and fp6 = e3m2 — the same element formats gfx950's
MX-scaled MFMA consumes — fp8 = e4m3fn (hardware convert), fp12 = e5m6.

GPU path: HIP kernels (csrc/quantize.hip, enc_fp_em/dec_fp_em). CPU path:
the identical codec written against torch.frexp, used for tests and as a
host fallback. Compute containers are byte-aligned u16; pack_codes /
unpack_codes provide the dense bitstream wire format for storage and
comms (4 bits -> 2/byte, 6 bits -> 4/3B, 12 bits -> 2/3B).
"""
import torch

from .loader import get_ext, has_ext

_FMT = {4: (2, 1), 6: (3, 2), 12: (5, 6)}


def _fmt_max(E, M):
    bias = (1 << (E - 1)) - 1
    return (2.0 - 1.0 / (1 << M)) * 2.0 ** (((1 << E) - 1) - bias)


def _encode_torch(x, E, M):
    """Round fp32 -> (E,M) float, returned as the decoded fp32 value and
    the raw bit pattern. Saturating, RNE, subnormal-correct."""
    bias = (1 << (E - 1)) - 1
    maxval = _fmt_max(E, M)
    s = (x < 0).to(torch.int32)
    a = x.abs().nan_to_num(nan=maxval).clamp(max=maxval)
    m, e = torch.frexp(a)          # a = m * 2^e, m in [0.5, 1)
    ebits = e - 1 + bias
    normal = ebits >= 1
    # normal: q = round((2m-1)*2^M)
    qn = torch.round((2 * m - 1) * (1 << M)).to(torch.int32)
    carry = qn == (1 << M)
    qn = torch.where(carry, torch.zeros_like(qn), qn)
    ebits = torch.where(carry, ebits + 1, ebits)
    sat = ebits > ((1 << E) - 1)
    ebits = torch.where(sat, torch.full_like(ebits, (1 << E) - 1), ebits)
    qn = torch.where(sat, torch.full_like(qn, (1 << M) - 1), qn)
    # subnormal: q = round(a * 2^(bias-1+M))
    qs = torch.round(a * 2.0 ** (bias - 1 + M)).to(torch.int32)
    to_norm = qs >= (1 << M)
    eb_sub = torch.where(to_norm, torch.ones_like(qs), torch.zeros_like(qs))
    qs = torch.where(to_norm, torch.zeros_like(qs), qs)
    ebits = torch.where(normal, ebits, eb_sub)
    q = torch.where(normal, qn, qs)
    zero = a == 0
    ebits = torch.where(zero, torch.zeros_like(ebits), ebits)
    q = torch.where(zero, torch.zeros_like(q), q)
    bits = (s << (E + M)) | (ebits << M) | q
    return bits.to(torch.uint16)


def _decode_torch(bits, E, M):
    bias = (1 << (E - 1)) - 1
    b = bits.to(torch.int32)
    s = (b >> (E + M)) & 1
    ef = (b >> M) & ((1 << E) - 1)
    mf = (b & ((1 << M) - 1)).float()
    sub = ef == 0
    v = torch.where(
        sub, mf * 2.0 ** (1 - bias - M),
        (1.0 + mf / (1 << M)) * torch.pow(2.0, (ef - bias).float()))
    return torch.where(s.bool(), -v, v)


class FP_Quantize:
    def __init__(self, group_size=512, q_bits=8):
        self.group_size = group_size
        self.q_bits = q_bits
        self.orig_shape = None

    def quantize(self, input, q_bits=None, return_meta_tensor=False):
        q_bits = q_bits or self.q_bits
        self.orig_shape = tuple(input.shape)
        x = input.contiguous()
        if q_bits == 8:
            from .quantizer import quantize_fp8
            q, scales = quantize_fp8(x.to(torch.bfloat16).reshape(-1),
                                     self.group_size)
            self.scales = scales
            return (q, scales) if return_meta_tensor else q
        E, M = _FMT[q_bits]
        if x.is_cuda and has_ext():
            q, scales = get_ext().quantize_fp_em(
                x.to(torch.bfloat16).reshape(-1), q_bits, self.group_size)
        else:
            flat = x.float().reshape(-1)
            n = flat.numel()
            pad = (-n) % self.group_size
            g = torch.nn.functional.pad(flat, (0, pad)) \
                .reshape(-1, self.group_size)
            amax = g.abs().amax(dim=1)
            scales = torch.where(amax > 0, amax / _fmt_max(E, M),
                                 torch.ones_like(amax))
            q = _encode_torch(g / scales.unsqueeze(1), E, M) \
                .reshape(-1)[:n]
        self.scales = scales
        return (q, scales) if return_meta_tensor else q

    def dequantize(self, input_q, fp_out=None, q_bits=None, scale=None):
        q_bits = q_bits or self.q_bits
        scales = scale if scale is not None else self.scales
        if q_bits == 8:
            from .quantizer import dequantize_fp8
            out = dequantize_fp8(input_q, scales, self.group_size) \
                .reshape(self.orig_shape)
        else:
            E, M = _FMT[q_bits]
            if input_q.is_cuda and has_ext():
                out = get_ext().dequantize_fp_em(
                    input_q, scales, q_bits, self.group_size,
                    list(self.orig_shape))
            else:
                v = _decode_torch(input_q, E, M)
                n = v.numel()
                pad = (-n) % self.group_size
                v = torch.nn.functional.pad(v, (0, pad)) \
                    .reshape(-1, self.group_size)
                v = (v * scales.unsqueeze(1)).reshape(-1)[:n]
                out = v.reshape(self.orig_shape)
        if fp_out is not None:
            fp_out.copy_(out.to(fp_out.dtype))
            return fp_out
        return out


# ---------------------------------------------------------------- packing
# Dense bit-packed wire format for the 4/6/12-bit codes (u16 working
# containers stay for compute; packing is for storage/comms). Little-
# endian bitstream, vectorized with tensor shifts — runs on CPU or GPU.

def pack_codes(codes, bits):
    """uint16 codes [n] -> uint8 bitstream [ceil(n*bits/8)]."""
    assert bits in (4, 6, 12)
    c = codes.to(torch.int32)
    n = c.numel()
    if bits == 4:
        pad = (-n) % 2
        if pad:
            c = torch.nn.functional.pad(c, (0, pad))
        c = c.view(-1, 2)
        return (c[:, 0] | (c[:, 1] << 4)).to(torch.uint8)
    if bits == 6:  # 4 codes -> 3 bytes
        pad = (-n) % 4
        if pad:
            c = torch.nn.functional.pad(c, (0, pad))
        c = c.view(-1, 4)
        w = c[:, 0] | (c[:, 1] << 6) | (c[:, 2] << 12) | (c[:, 3] << 18)
        out = torch.stack([w & 0xFF, (w >> 8) & 0xFF, (w >> 16) & 0xFF],
                          dim=1)
        return out.reshape(-1).to(torch.uint8)
    # 12: 2 codes -> 3 bytes
    pad = (-n) % 2
    if pad:
        c = torch.nn.functional.pad(c, (0, pad))
    c = c.view(-1, 2)
    w = c[:, 0] | (c[:, 1] << 12)
    out = torch.stack([w & 0xFF, (w >> 8) & 0xFF, (w >> 16) & 0xFF], dim=1)
    return out.reshape(-1).to(torch.uint8)


def unpack_codes(packed, bits, n):
    """uint8 bitstream -> uint16 codes [n]."""
    assert bits in (4, 6, 12)
    b = packed.to(torch.int32)
    if bits == 4:
        lo = b & 0xF
        hi = (b >> 4) & 0xF
        return torch.stack([lo, hi], 1).reshape(-1)[:n].to(torch.uint16)
    if bits == 6:
        b = b.view(-1, 3)
        w = b[:, 0] | (b[:, 1] << 8) | (b[:, 2] << 16)
        c = torch.stack([w & 0x3F, (w >> 6) & 0x3F, (w >> 12) & 0x3F,
                         (w >> 18) & 0x3F], dim=1)
        return c.reshape(-1)[:n].to(torch.uint16)
    b = b.view(-1, 3)
    w = b[:, 0] | (b[:, 1] << 8) | (b[:, 2] << 16)
    c = torch.stack([w & 0xFFF, (w >> 12) & 0xFFF], dim=1)
    return c.reshape(-1)[:n].to(torch.uint16)
