"""Runtime helpers: flattening, grad-norm, overflow checks, memory reports.

Parity: reference `deepspeed/runtime/utils.py` (`clip_grad_norm_:359`,
`see_memory_usage:815`, CheckOverflow, partition helpers).
"""
import math

import torch

from .. import comm as dist
from ..utils.logging import logger


def noop_decorator(fn):
    return fn


# -- flatten/unflatten (apex-style, via torch C util) ------------------------

def flatten_dense_tensors(tensors):
    return torch._C._nn.flatten_dense_tensors(tensors)


def unflatten_dense_tensors(flat, tensors):
    return torch._C._nn.unflatten_dense_tensors(flat, tensors)


def align_dense_tensors(tensor_list, alignment):
    """Pad the final tensor so the total numel is a multiple of alignment."""
    num_elements = sum(t.numel() for t in tensor_list)
    remaining = num_elements % alignment
    if remaining == 0:
        return tensor_list
    pad = alignment - remaining
    pad_tensor = torch.zeros(pad, device=tensor_list[0].device,
                             dtype=tensor_list[0].dtype)
    return tensor_list + [pad_tensor]


def get_grad_norm(parameters, mpu=None, norm_type=2.0, group=None):
    """Total grad norm across the data-parallel group (and MP group if set)."""
    if isinstance(parameters, torch.Tensor):
        parameters = [parameters]
    grads = [p.grad for p in parameters if p.grad is not None]
    return get_flat_grad_norm(grads, norm_type=norm_type, group=group)


def get_flat_grad_norm(grads, norm_type=2.0, group=None):
    norm_type = float(norm_type)
    if norm_type == math.inf:
        total = max((g.abs().max().item() for g in grads), default=0.0)
        if dist.is_initialized():
            t = torch.tensor([total], device=grads[0].device if grads else "cpu",
                             dtype=torch.float32)
            dist.all_reduce(t, op=dist.ReduceOp.MAX, group=group)
            total = t.item()
        return total
    total = 0.0
    for g in grads:
        total += float(g.float().norm(norm_type)**norm_type)
    if dist.is_initialized():
        device = grads[0].device if grads else "cpu"
        t = torch.tensor([total], device=device, dtype=torch.float64)
        dist.all_reduce(t, group=group)
        total = t.item()
    return total**(1.0 / norm_type)


def clip_tensors_by_global_norm(tensors, max_norm, global_norm, eps=1e-6):
    clip_coef = max_norm / (global_norm + eps)
    if clip_coef < 1.0:
        for t in tensors:
            t.mul_(clip_coef)
    return global_norm


def clip_grad_norm_(parameters, max_norm, norm_type=2, mpu=None,
                    group=None):
    """Clip the gradients of `parameters` by their global norm across
    ranks (reference-compatible name, ref runtime/utils.py:359)."""
    if isinstance(parameters, torch.Tensor):
        parameters = [parameters]
    grads = [p.grad for p in parameters if p.grad is not None]
    if not grads:
        return 0.0
    total = get_flat_grad_norm(grads, norm_type=norm_type, group=group)
    clip_tensors_by_global_norm(grads, max_norm, total)
    return total


class CheckOverflow:
    """Detect inf/nan in gradients across ranks (fp16 dynamic loss scaling)."""

    def __init__(self, param_groups=None, mpu=None, deepspeed=None):
        self.params = []
        if param_groups:
            for group in param_groups:
                self.params.extend(group)

    @staticmethod
    def _has_inf_or_nan(x):
        try:
            s = float(x.float().sum())
        except RuntimeError:
            return True
        return s in (float("inf"), float("-inf")) or s != s

    def check(self, param_groups=None):
        params = []
        if param_groups is None:
            params = self.params
        else:
            for g in param_groups:
                params.extend(g)
        return self.has_overflow([p.grad for p in params if p.grad is not None])

    def has_overflow(self, grads, group=None):
        overflow = any(self._has_inf_or_nan(g) for g in grads)
        if dist.is_initialized():
            device = grads[0].device if grads else "cpu"
            t = torch.tensor([1.0 if overflow else 0.0], device=device)
            dist.all_reduce(t, op=dist.ReduceOp.MAX, group=group)
            overflow = bool(t.item())
        return overflow


def see_memory_usage(message, force=False, ranks=[0]):
    if not force:
        return
    if dist.is_initialized() and dist.get_rank() not in ranks:
        return
    import gc
    import psutil
    gc.collect()
    vm = psutil.virtual_memory()
    msg = f"{message} | CPU used {vm.used/2**30:.2f}GB ({vm.percent}%)"
    if torch.cuda.is_available():
        msg += (f" | HBM alloc {torch.cuda.memory_allocated()/2**30:.2f}GB "
                f"max {torch.cuda.max_memory_allocated()/2**30:.2f}GB "
                f"reserved {torch.cuda.memory_reserved()/2**30:.2f}GB")
        torch.cuda.reset_peak_memory_stats()
    logger.info(msg)


def empty_cache():
    if torch.cuda.is_available():
        torch.cuda.empty_cache()


def partition_uniform(num_items, num_parts):
    """Balanced contiguous partition bounds: len == num_parts+1."""
    parts = [0] * (num_parts + 1)
    chunk = num_items // num_parts
    rem = num_items % num_parts
    for p in range(num_parts):
        parts[p + 1] = parts[p] + chunk + (1 if p < rem else 0)
    return parts


def partition_balanced(weights, num_parts):
    """Partition items with weights into contiguous chunks minimizing the max
    chunk weight (binary search + greedy check). Returns bounds len
    num_parts+1."""
    n = len(weights)
    if n <= num_parts:
        return partition_uniform(n, num_parts)
    prefix = [0]
    for w in weights:
        prefix.append(prefix[-1] + w)

    def can(limit):
        bounds = [0]
        cur = 0
        for i in range(1, num_parts + 1):
            # furthest j with prefix[j]-prefix[cur] <= limit
            lo, hi = cur, n
            while lo < hi:
                mid = (lo + hi + 1) // 2
                if prefix[mid] - prefix[cur] <= limit:
                    lo = mid
                else:
                    hi = mid - 1
            if lo == cur and cur < n:
                return None  # single item exceeds limit
            bounds.append(lo)
            cur = lo
        return bounds if bounds[-1] == n else None

    lo = max(weights) if weights else 0
    hi = prefix[-1]
    best = None
    while lo <= hi:
        mid = (lo + hi) // 2
        b = can(mid)
        if b is not None:
            best = b
            hi = mid - 1
        else:
            lo = mid + 1
    if best is None:
        return partition_uniform(n, num_parts)
    # ensure monotone coverage
    best[-1] = n
    return best


def call_to_str(base, *args, **kwargs):
    """Pretty call repr for pipeline instruction logging."""
    name = f"{base}("
    if args:
        name += ", ".join(repr(a) for a in args)
        if kwargs:
            name += ", "
    if kwargs:
        name += ", ".join(f"{k}={v!r}" for k, v in kwargs.items())
    name += ")"
    return name


def assert_ints_same_as_other_ranks(ints, group=None, tag=""):
    """Debug-mode cross-rank agreement check (ref stage3.py:1594).

    Collectives deadlock or silently corrupt when ranks disagree on the
    order/identity of the tensors they reduce; this asserts the given
    list of ints (e.g. param ids in a reduction bucket) is identical on
    every rank of the group. Enable via DSAMD_SANITY=1 — it costs one
    small all_gather per call, cheap insurance on a 7-link async fabric.
    """
    from .. import comm as dist
    if not dist.is_initialized():
        return
    world = dist.get_world_size(group)
    if world <= 1:
        return
    t = torch.tensor(list(ints), dtype=torch.long)
    if torch.cuda.is_available():
        t = t.cuda()
    sizes = [torch.zeros(1, dtype=torch.long, device=t.device)
             for _ in range(world)]
    dist.all_gather(sizes, torch.tensor([t.numel()], dtype=torch.long,
                                        device=t.device), group=group)
    assert all(s.item() == t.numel() for s in sizes),         f"[sanity:{tag}] rank list LENGTHS differ: "         f"{[s.item() for s in sizes]}"
    gathered = [torch.zeros_like(t) for _ in range(world)]
    dist.all_gather(gathered, t, group=group)
    for r, g in enumerate(gathered):
        assert torch.equal(g, gathered[0]),             f"[sanity:{tag}] rank {r} disagrees with rank 0 on ids"
