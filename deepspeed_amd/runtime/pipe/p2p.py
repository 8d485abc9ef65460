"""Point-to-point activation/grad exchange between adjacent stages.

Parity: reference `runtime/pipe/p2p.py:46,67` + meta handshake
(`pipe/engine.py:956`). Sends are async (isend) so opposing send/send pairs
between neighbors can never deadlock; receives are blocking.
"""
import torch

from ... import comm as dist

_DTYPE_CODES = {
    torch.float32: 0, torch.float16: 1, torch.bfloat16: 2, torch.int64: 3,
    torch.int32: 4, torch.bool: 5,
}
_CODE_DTYPES = {v: k for k, v in _DTYPE_CODES.items()}
MAX_DIMS = 8

_pending = []  # in-flight isend works (drained at batch end)


def _meta_tensor(tensor):
    meta = torch.zeros(2 + MAX_DIMS, dtype=torch.long)
    meta[0] = _DTYPE_CODES[tensor.dtype]
    meta[1] = tensor.dim()
    for i, s in enumerate(tensor.shape):
        meta[2 + i] = s
    return meta


def send_meta(tensor, dst_rank):
    dist.send(_meta_tensor(tensor), dst_rank)


def recv_meta(src_rank):
    meta = torch.zeros(2 + MAX_DIMS, dtype=torch.long)
    dist.recv(meta, src_rank)
    dtype = _CODE_DTYPES[int(meta[0])]
    shape = [int(meta[2 + i]) for i in range(int(meta[1]))]
    return dtype, shape


def isend(tensor, dst_rank):
    work = dist.isend(tensor.contiguous(), dst_rank)
    if work is not None:
        _pending.append((work, tensor))
    return work


def recv(tensor, src_rank):
    dist.recv(tensor, src_rank)
    return tensor


def drain():
    """Wait for all outstanding sends (call at batch boundary)."""
    global _pending
    for work, _ in _pending:
        work.wait()
    _pending = []
