"""Domino: tensor parallelism with communication hidden behind compute.

Parity: reference `runtime/domino/transformer.py` (DominoTransformerLayer) /
`async_linear.py`. The row-parallel all-reduce is split into micro-chunks:
chunk i's all-reduce runs asynchronously on the wire while chunk i+1's GEMM
executes — on fully-connected xGMI the all-reduce streams over all 7 links
concurrently with hipBLASLt on the CUs.
"""
import torch
import torch.nn.functional as F

from .. import comm as dist
from ..module_inject.layers import _ReduceFromTensorParallel


class _DominoRowParallelFn(torch.autograd.Function):
    """Forward: chunked GEMM + async all-reduce overlap. Backward: plain
    (grad wrt input is local: dy @ W)."""

    @staticmethod
    def forward(ctx, x, weight, group, n_chunks):
        ctx.save_for_backward(x, weight)
        ctx.group = group
        world = dist.get_world_size(group)
        chunks = x.tensor_split(n_chunks, dim=0)
        outs = []
        works = []
        for c in chunks:
            y = F.linear(c, weight)
            if world > 1:
                w = dist.all_reduce(y, group=group, async_op=True)
                works.append(w)
            outs.append(y)
        for w in works:
            if w is not None:
                w.wait()
        return torch.cat(outs, dim=0)

    @staticmethod
    def backward(ctx, dy):
        x, weight = ctx.saved_tensors
        dx = dy @ weight          # local (input was sharded)
        dw = dy.reshape(-1, dy.shape[-1]).T @ x.reshape(-1, x.shape[-1])
        return dx, dw, None, None


class DominoLinearAllreduce(torch.nn.Module):
    """Row-parallel linear with chunked async all-reduce overlap."""

    def __init__(self, weight_shard, bias=None, group=None, n_chunks=2):
        super().__init__()
        self.weight = torch.nn.Parameter(weight_shard)
        self.bias = torch.nn.Parameter(bias) if bias is not None else None
        self.group = group
        self.n_chunks = n_chunks

    @classmethod
    def from_linear(cls, linear, group, rank, world, n_chunks=2):
        infe = linear.in_features
        assert infe % world == 0
        chunk = infe // world
        w = linear.weight.data[:, rank * chunk:(rank + 1) * chunk].clone()
        b = linear.bias.data.clone() if linear.bias is not None else None
        return cls(w, b, group, n_chunks)

    def forward(self, x):
        orig_shape = x.shape
        flat = x.reshape(-1, x.shape[-1])
        if flat.shape[0] >= self.n_chunks and self.training:
            y = _DominoRowParallelFn.apply(flat, self.weight, self.group,
                                           self.n_chunks)
        else:
            y = F.linear(flat, self.weight)
            y = _ReduceFromTensorParallel.apply(y, self.group)
        y = y.reshape(*orig_shape[:-1], self.weight.shape[0])
        if self.bias is not None:
            y = y + self.bias
        return y


def enable_domino(model, tp_group=None, n_chunks=2):
    """Swap row-parallel LinearAllreduce layers for Domino variants."""
    from ..module_inject.layers import LinearAllreduce
    targets = []
    for parent in model.modules():
        for name, child in parent.named_children():
            if type(child) is LinearAllreduce:
                targets.append((parent, name, child))
    for parent, name, child in targets:
        dom = DominoLinearAllreduce(child.weight.data,
                                    child.bias.data
                                    if child.bias is not None else None,
                                    child.group, n_chunks)
        setattr(parent, name, dom)
    return model
