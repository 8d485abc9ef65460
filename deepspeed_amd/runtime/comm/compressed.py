"""Error-feedback 1-bit compressed allreduce.

Parity: reference `runtime/comm/nccl.py` (NcclBackend.compressed_allreduce)
/ `runtime/comm/compressed.py:14`. Two phases over the EP-friendly xGMI
fabric: all-to-all of sign+scale compressed worker chunks, local decompress/
average/recompress, then all-gather of the server chunks. Error feedback
keeps the quantization bias out of convergence.
"""
import torch

from ... import comm as dist

_POW2 = None


def _pow2(device):
    global _POW2
    if _POW2 is None or _POW2.device != device:
        _POW2 = torch.tensor([1, 2, 4, 8, 16, 32, 64, 128],
                             dtype=torch.uint8, device=device)
    return _POW2


def _pack_signs(x):
    """x [n] -> uint8 [n/8] of sign bits (1 = non-negative)."""
    bits = (x >= 0).to(torch.uint8).reshape(-1, 8)
    return (bits * _pow2(x.device)).sum(dim=1, dtype=torch.uint8)


def _unpack_signs(packed, n):
    bits = packed.unsqueeze(1).bitwise_and(
        _pow2(packed.device).unsqueeze(0)).ne(0)
    out = bits.reshape(-1)[:n].to(torch.float32)
    return out * 2.0 - 1.0  # {0,1} -> {-1,+1}


def _compress(x):
    """Returns (packed_signs uint8, scale fp32 scalar-tensor, decompressed)."""
    n = x.numel()
    pad = (-n) % 8
    if pad:
        x = torch.nn.functional.pad(x, (0, pad))
    scale = x.abs().mean()
    packed = _pack_signs(x)
    decompressed = _unpack_signs(packed, n) * scale
    return packed, scale, decompressed


class CompressedBackend:
    """Torch-collective implementation (works over RCCL and gloo)."""

    def __init__(self, mpu=None):
        pass

    def compressed_allreduce(self, buffer_m, worker_error, server_error,
                             local_rank=None, group=None):
        """In-place 1-bit allreduce of buffer_m with error feedback."""
        world = dist.get_world_size(group)
        original_shape = buffer_m.shape
        original_numel = buffer_m.numel()
        flat = buffer_m.reshape(-1).float()
        # pad so each rank chunk is a multiple of 8
        chunk = (original_numel + world - 1) // world
        chunk = (chunk + 7) // 8 * 8
        padded_numel = chunk * world
        if padded_numel != original_numel:
            flat = torch.nn.functional.pad(
                flat, (0, padded_numel - original_numel))
        if worker_error.numel() != padded_numel:
            worker_error.resize_(padded_numel).zero_()
        if server_error.numel() != chunk:
            server_error.resize_(chunk).zero_()

        flat += worker_error
        # per-destination-chunk compression
        packed_list, scales, decomp = [], [], []
        for r in range(world):
            seg = flat[r * chunk:(r + 1) * chunk]
            p, s, d = _compress(seg)
            packed_list.append(p)
            scales.append(s)
            decomp.append(d)
        worker_error.copy_(flat - torch.cat(decomp))

        if world > 1:
            sign_in = torch.cat(packed_list)
            sign_out = torch.empty_like(sign_in)
            dist.all_to_all_single(sign_out, sign_in, group=group)
            scale_in = torch.stack(scales)
            scale_out = torch.empty_like(scale_in)
            dist.all_to_all_single(scale_out, scale_in, group=group)
        else:
            sign_out = packed_list[0]
            scale_out = torch.stack(scales)

        # server phase: decompress all workers' copies of my chunk, average
        per = chunk // 8
        server = torch.zeros(chunk, device=flat.device)
        for r in range(world):
            seg = sign_out[r * per:(r + 1) * per]
            server += _unpack_signs(seg, chunk) * scale_out[r]
        server /= world
        server += server_error
        sp, ss, sd = _compress(server)
        server_error.copy_(server - sd)

        if world > 1:
            gathered_signs = torch.empty(per * world, dtype=torch.uint8,
                                         device=flat.device)
            dist.all_gather_into_tensor(gathered_signs, sp, group=group)
            gathered_scales = torch.empty(world, device=flat.device)
            dist.all_gather_into_tensor(gathered_scales,
                                        ss.reshape(1), group=group)
        else:
            gathered_signs = sp
            gathered_scales = ss.reshape(1)

        out = torch.cat([
            _unpack_signs(gathered_signs[r * per:(r + 1) * per], chunk) *
            gathered_scales[r] for r in range(world)])
        buffer_m.copy_(out[:original_numel].reshape(original_shape)
                       .to(buffer_m.dtype))
        return buffer_m
