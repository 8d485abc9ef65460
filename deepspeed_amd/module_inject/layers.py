"""Tensor-parallel linear layers.

Parity: reference `module_inject/layers.py` (`LinearAllreduce:752`,
`LinearLayer:833`, autograd ops RowParallel:159 / ColumnParallel:220).

MI355X note: TP all-reduces ride intra-node xGMI; TP groups are adjacent
ranks (comm/groups.py) so each all-reduce spans the fully-connected node.
"""
import torch
import torch.nn.functional as F

from .. import comm as dist


class _CopyToTensorParallel(torch.autograd.Function):
    """Identity fwd; all-reduce grads in bwd (column-parallel input)."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return x

    @staticmethod
    def backward(ctx, grad):
        if dist.get_world_size(ctx.group) > 1:
            grad = grad.contiguous()
            dist.all_reduce(grad, group=ctx.group)
        return grad, None


class _ReduceFromTensorParallel(torch.autograd.Function):
    """All-reduce fwd; identity bwd (row-parallel output)."""

    @staticmethod
    def forward(ctx, x, group):
        if dist.get_world_size(group) > 1:
            x = x.contiguous()
            dist.all_reduce(x, group=group)
        return x

    @staticmethod
    def backward(ctx, grad):
        return grad, None


class LinearLayer(torch.nn.Module):
    """Column-parallel: weight rows sharded; output stays sharded."""

    def __init__(self, weight_shard, bias_shard=None, group=None):
        super().__init__()
        self.weight = torch.nn.Parameter(weight_shard)
        self.bias = (torch.nn.Parameter(bias_shard)
                     if bias_shard is not None else None)
        self.group = group

    @classmethod
    def from_linear(cls, linear, group, rank, world):
        out = linear.out_features
        assert out % world == 0, f"out_features {out} % tp {world} != 0"
        chunk = out // world
        w = linear.weight.data[rank * chunk:(rank + 1) * chunk].clone()
        b = (linear.bias.data[rank * chunk:(rank + 1) * chunk].clone()
             if linear.bias is not None else None)
        return cls(w, b, group)

    def forward(self, x):
        x = _CopyToTensorParallel.apply(x, self.group)
        return F.linear(x, self.weight, self.bias)


class LinearAllreduce(torch.nn.Module):
    """Row-parallel: weight cols sharded; output all-reduced."""

    def __init__(self, weight_shard, bias=None, group=None):
        super().__init__()
        self.weight = torch.nn.Parameter(weight_shard)
        self.bias = torch.nn.Parameter(bias) if bias is not None else None
        self.group = group

    @classmethod
    def from_linear(cls, linear, group, rank, world):
        infe = linear.in_features
        assert infe % world == 0, f"in_features {infe} % tp {world} != 0"
        chunk = infe // world
        w = linear.weight.data[:, rank * chunk:(rank + 1) * chunk].clone()
        # bias replicated, applied once after the reduce on every rank
        b = linear.bias.data.clone() if linear.bias is not None else None
        return cls(w, b, group)

    def forward(self, x):
        y = F.linear(x, self.weight)
        y = _ReduceFromTensorParallel.apply(y, self.group)
        if self.bias is not None:
            y = y + self.bias
        return y
