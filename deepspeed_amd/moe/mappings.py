"""MoE <-> tensor-parallel token mappings.

Parity: reference `deepspeed/moe/mappings.py` (gather_tokens /
drop_tokens). Under TP, activations are replicated across the TP group;
dropping tokens (each TP rank keeps its 1/tp slice of the sequence)
before the MoE all-to-all divides the dispatch volume by tp, and
gathering after restores the replicated layout.
"""
import torch

from .. import comm as dist
from ..comm import groups as grp


def _tp_group(group):
    return group if group is not None else grp.get_tensor_parallel_group()


class _GatherTokens(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, dim, group):
        ctx.dim = dim
        ctx.group = group
        world = dist.get_world_size(group)
        if world == 1:
            return x
        rank = dist.get_rank(group)
        ctx.rank = rank
        parts = [torch.empty_like(x) for _ in range(world)]
        dist.all_gather(parts, x.contiguous(), group=group)
        return torch.cat(parts, dim=dim)

    @staticmethod
    def backward(ctx, grad):
        world = dist.get_world_size(ctx.group)
        if world == 1:
            return grad, None, None
        return (grad.chunk(world, dim=ctx.dim)[ctx.rank].contiguous(),
                None, None)


class _DropTokens(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, dim, group):
        ctx.dim = dim
        ctx.group = group
        world = dist.get_world_size(group)
        if world == 1:
            return x
        rank = dist.get_rank(group)
        return x.chunk(world, dim=dim)[rank].contiguous()

    @staticmethod
    def backward(ctx, grad):
        world = dist.get_world_size(ctx.group)
        if world == 1:
            return grad, None, None
        parts = [torch.empty_like(grad) for _ in range(world)]
        dist.all_gather(parts, grad.contiguous(), group=ctx.group)
        return torch.cat(parts, dim=ctx.dim), None, None


def gather_tokens(x, dim=1, group=None):
    """Rebuild the full (replicated) sequence from per-TP-rank slices."""
    g = _tp_group(group)
    if g is None:  # no tensor parallelism configured: identity
        return x
    return _GatherTokens.apply(x, dim, g)


def drop_tokens(x, dim=1, group=None):
    """Keep this TP rank's 1/tp slice of the sequence (inverse of
    gather_tokens); sequence length must divide the TP degree."""
    g = _tp_group(group)
    if g is None:
        return x
    assert x.shape[dim] % dist.get_world_size(g) == 0, \
        f"dim {dim} size {x.shape[dim]} not divisible by tp"
    return _DropTokens.apply(x, dim, g)
