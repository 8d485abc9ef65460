"""Ulysses sequence parallelism: head-scatter all-to-all attention.

Parity: reference `deepspeed/sequence/layer.py:351` (DistributedAttention),
`:241` (single_all_to_all), `:297` (_SeqAllToAll).

MI355X note: all-to-all is the *best-matched* collective for a
fully-connected 8-GPU xGMI node — each rank streams to all 7 peers
simultaneously, so Ulysses' 4 a2a per attention ride the full 7x153 GB/s
fan-out rather than a per-link-bound ring.
"""
import torch

from .. import comm as dist
from ..comm import groups


def single_all_to_all(input_, scatter_idx, gather_idx, group):
    """All-to-all that scatters dim `scatter_idx` and gathers `gather_idx`.

    Training layouts:
      fwd qkv: [B, s/P, H, D]  -(scatter 2, gather 1)->  [B, s, H/P, D]
      fwd out: [B, s, H/P, D]  -(scatter 1, gather 2)->  [B, s/P, H, D]
    """
    world = dist.get_world_size(group)
    if world == 1:
        return input_
    B = input_.shape[0]
    if scatter_idx == 2 and gather_idx == 1:
        s, H, D = input_.shape[1], input_.shape[2], input_.shape[3]
        assert H % world == 0, f"heads {H} % sp {world} != 0"
        hp = H // world
        # [B, s, P, hp, D] -> [P, s, B, hp, D]
        t = input_.reshape(B, s, world, hp, D).permute(2, 1, 0, 3, 4) \
            .contiguous()
        out = torch.empty_like(t)
        dist.all_to_all_single(out, t, group=group)
        # out: [P(seq chunk), s, B, hp, D] -> [B, P*s, hp, D]
        return out.permute(2, 0, 1, 3, 4).reshape(B, world * s, hp, D)
    elif scatter_idx == 1 and gather_idx == 2:
        S, hp, D = input_.shape[1], input_.shape[2], input_.shape[3]
        assert S % world == 0
        s = S // world
        # [B, P, s, hp, D] -> [P, s, B, hp, D]
        t = input_.reshape(B, world, s, hp, D).permute(1, 2, 0, 3, 4) \
            .contiguous()
        out = torch.empty_like(t)
        dist.all_to_all_single(out, t, group=group)
        # out: [P(head chunk), s, B, hp, D] -> [B, s, P*hp, D]
        return out.permute(2, 1, 0, 3, 4).reshape(B, s, world * hp, D)
    raise ValueError(f"unsupported scatter {scatter_idx} gather {gather_idx}")


class _SeqAllToAll(torch.autograd.Function):
    @staticmethod
    def forward(ctx, group, input_, scatter_idx, gather_idx):
        ctx.group = group
        ctx.scatter_idx = scatter_idx
        ctx.gather_idx = gather_idx
        return single_all_to_all(input_, scatter_idx, gather_idx, group)

    @staticmethod
    def backward(ctx, grad):
        return (None,
                single_all_to_all(grad.contiguous(), ctx.gather_idx,
                                  ctx.scatter_idx, ctx.group),
                None, None)


class DistributedAttention(torch.nn.Module):
    """Wraps a local attention fn: seq-sharded in, seq-sharded out.

    local_attn(q, k, v, *args, **kwargs) operates on [B, S_full, H/P, D].
    """

    def __init__(self, local_attention, sequence_process_group=None,
                 scatter_idx=2, gather_idx=1):
        super().__init__()
        self.local_attn = local_attention
        self.spg = sequence_process_group
        self.scatter_idx = scatter_idx
        self.gather_idx = gather_idx

    def forward(self, query, key, value, *args, **kwargs):
        spg = self.spg if self.spg is not None \
            else groups.get_sequence_parallel_group()
        q = _SeqAllToAll.apply(spg, query, self.scatter_idx, self.gather_idx)
        k = _SeqAllToAll.apply(spg, key, self.scatter_idx, self.gather_idx)
        v = _SeqAllToAll.apply(spg, value, self.scatter_idx, self.gather_idx)
        ctx = self.local_attn(q, k, v, *args, **kwargs)
        return _SeqAllToAll.apply(spg, ctx, self.gather_idx, self.scatter_idx)
