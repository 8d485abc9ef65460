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


def _head_counts(H, world):
    """Per-rank head counts for uneven head distribution (ref
    sequence/layer.py:131): first H%world ranks take one extra head."""
    base, extra = divmod(H, world)
    return [base + (1 if r < extra else 0) for r in range(world)]


def single_all_to_all(input_, scatter_idx, gather_idx, group,
                      uneven_heads=None):
    """All-to-all that scatters dim `scatter_idx` and gathers `gather_idx`.

    Training layouts:
      fwd qkv: [B, s/P, H, D]  -(scatter 2, gather 1)->  [B, s, H/P, D]
      fwd out: [B, s, H/P, D]  -(scatter 1, gather 2)->  [B, s/P, H, D]
    Head counts that do not divide the SP degree distribute unevenly
    (first ranks take one extra head) via unequal a2a splits.
    """
    world = dist.get_world_size(group)
    if world == 1:
        return input_
    B = input_.shape[0]
    if scatter_idx == 2 and gather_idx == 1:
        s, H, D = input_.shape[1], input_.shape[2], input_.shape[3]
        if H % world != 0:
            return _a2a_uneven_fwd(input_, group, world)  # noqa: E501 (uneven)
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
        if uneven_heads is not None:
            return _a2a_uneven_bwd(input_, group, world, uneven_heads)
        # [B, P, s, hp, D] -> [P, s, B, hp, D]
        t = input_.reshape(B, world, s, hp, D).permute(1, 2, 0, 3, 4) \
            .contiguous()
        out = torch.empty_like(t)
        dist.all_to_all_single(out, t, group=group)
        # out: [P(head chunk), s, B, hp, D] -> [B, s, P*hp, D]
        return out.permute(2, 1, 0, 3, 4).reshape(B, s, world * hp, D)
    raise ValueError(f"unsupported scatter {scatter_idx} gather {gather_idx}")


def _a2a_uneven_fwd(input_, group, world):
    """[B, s, H, D] -> [B, s*P, hc(rank), D] with uneven head counts."""
    B, s, H, D = input_.shape
    counts = _head_counts(H, world)
    rank = dist.get_rank(group)
    # send rank r its head slice (for ALL of my seq chunk)
    t = input_.permute(2, 1, 0, 3).contiguous()    # [H, s, B, D]
    in_splits = [c * s * B * D for c in counts]
    out_splits = [counts[rank] * s * B * D] * world
    out = torch.empty(world * counts[rank] * s * B * D, dtype=t.dtype,
                      device=t.device)
    dist.all_to_all_single(out, t.reshape(-1),
                           output_split_sizes=out_splits,
                           input_split_sizes=in_splits, group=group)
    # out: per source rank [hc(rank), s, B, D] -> [B, P*s, hc, D]
    o = out.reshape(world, counts[rank], s, B, D)         .permute(3, 0, 2, 1, 4).reshape(B, world * s, counts[rank], D)
    return o.contiguous()


def _a2a_uneven_bwd(input_, group, world, H):
    """[B, S, hc(rank), D] -> [B, S/P, H, D] (inverse of uneven fwd)."""
    B, S, hp, D = input_.shape
    s = S // world
    counts = _head_counts(H, world)
    rank = dist.get_rank(group)
    # [B, P, s, hp, D] -> [P, hp, s, B, D]
    t = input_.reshape(B, world, s, hp, D).permute(1, 3, 2, 0, 4)         .contiguous()
    in_splits = [hp * s * B * D] * world
    out_splits = [c * s * B * D for c in counts]
    out = torch.empty(sum(out_splits), dtype=t.dtype, device=t.device)
    dist.all_to_all_single(out, t.reshape(-1),
                           output_split_sizes=out_splits,
                           input_split_sizes=in_splits, group=group)
    # concat head chunks [hc(r), s, B, D] over r -> [H, s, B, D]
    pieces = []
    off = 0
    for c in counts:
        n = c * s * B * D
        pieces.append(out[off:off + n].reshape(c, s, B, D))
        off += n
    o = torch.cat(pieces, dim=0).permute(2, 1, 0, 3)  # [B, s, H, D]
    return o.contiguous()


class _SeqAllToAll(torch.autograd.Function):
    @staticmethod
    def forward(ctx, group, input_, scatter_idx, gather_idx,
                uneven_heads=None):
        ctx.group = group
        ctx.scatter_idx = scatter_idx
        ctx.gather_idx = gather_idx
        ctx.uneven_heads = uneven_heads
        return single_all_to_all(input_, scatter_idx, gather_idx, group,
                                 uneven_heads)

    @staticmethod
    def backward(ctx, grad):
        return (None,
                single_all_to_all(grad.contiguous(), ctx.gather_idx,
                                  ctx.scatter_idx, ctx.group,
                                  ctx.uneven_heads),
                None, None, None)


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
        world = dist.get_world_size(spg)
        hq = query.shape[2]
        hk = key.shape[2]
        if hk != hq and hk % world != 0 and world % hk == 0 \
                and hq % world == 0:
            # SP degree beyond the KV-head count (GQA/MQA): replicate KV
            # heads so every rank holds whole groups; autograd sums the
            # replica grads back (ref layer.py uneven/kv-replication path)
            rep = world // hk
            key = key.repeat_interleave(rep, dim=2)
            value = value.repeat_interleave(rep, dim=2)
            hk = key.shape[2]
        uneven_q = hq if hq % world != 0 else None
        uneven_k = hk if hk % world != 0 else None
        if uneven_q or uneven_k:
            assert hq == hk, \
                "uneven-head SP requires MHA (GQA needs divisible kv heads)"
        q = _SeqAllToAll.apply(spg, query, self.scatter_idx,
                               self.gather_idx, uneven_q)
        k = _SeqAllToAll.apply(spg, key, self.scatter_idx,
                               self.gather_idx, uneven_k)
        v = _SeqAllToAll.apply(spg, value, self.scatter_idx,
                               self.gather_idx, uneven_k)
        ctx = self.local_attn(q, k, v, *args, **kwargs)
        return _SeqAllToAll.apply(spg, ctx, self.gather_idx,
                                  self.scatter_idx, uneven_q)
