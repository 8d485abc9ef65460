"""Ulysses-SP / ALST adapter for HuggingFace transformers models.

Parity: reference `runtime/sequence_parallel/ulysses_sp.py` (UlyssesSP
attention adapter + sequence-sharded batches + tiled loss — "Arctic Long
Sequence Training"). This registers a `"ulysses_sp"` attention
implementation with transformers' AttentionInterface: q/k/v arrive
seq-sharded `[B, H, s/P, D]`, are head-scattered/seq-gathered with the
same `_SeqAllToAll` autograd op the native models use (4 all-to-alls per
attention — the best-matched collective for fully-connected xGMI), run
through torch SDPA at full sequence length, and return seq-sharded
output.

Restrictions (ALST's packed/causal regime): full causal attention, no
padding mask — the adapter asserts `attention_mask is None`. Labels must
be shifted BEFORE sharding (the causal shift crosses shard boundaries):
use `shard_batch_for_sp`.
"""
import torch
import torch.nn.functional as F

from ..comm import groups
from ..sequence.layer import _SeqAllToAll
from ..utils.logging import log_dist

_IMPL_NAME = "ulysses_sp"
_SP_GROUP = [None]


def _ulysses_attention(module, query, key, value, attention_mask,
                       scaling=None, dropout=0.0, **kwargs):
    """transformers AttentionInterface entry.

    query/key/value: [B, H, s_local, D]; returns ([B, s_local, H, D], None).
    """
    assert attention_mask is None, \
        "ulysses_sp supports packed/causal batches only (no padding mask)"
    spg = _SP_GROUP[0]
    if spg is None and groups.sequence_parallel_is_initialized():
        spg = groups.get_sequence_parallel_group()
    # [B, H, s, D] -> [B, s, H, D] for the a2a layout
    q = query.transpose(1, 2).contiguous()
    k = key.transpose(1, 2).contiguous()
    v = value.transpose(1, 2).contiguous()
    q = _SeqAllToAll.apply(spg, q, 2, 1)   # [B, S, H/P, D]
    k = _SeqAllToAll.apply(spg, k, 2, 1)
    v = _SeqAllToAll.apply(spg, v, 2, 1)
    out = F.scaled_dot_product_attention(
        q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
        dropout_p=dropout, scale=scaling, is_causal=True,
        enable_gqa=(k.shape[2] != q.shape[2]))
    out = out.transpose(1, 2).contiguous()          # [B, S, H/P, D]
    out = _SeqAllToAll.apply(spg, out, 1, 2)        # [B, s, H, D]
    return out, None


def apply_ulysses_sp_to_hf(model, sp_group=None, sp_size=None):
    """Switch an HF model's attention to the Ulysses implementation.

    Builds the SP process group when `sp_size` is given; KV-head count
    must be divisible by the SP degree."""
    if sp_group is None:
        if sp_size is not None and \
                not groups.sequence_parallel_is_initialized():
            groups.initialize_sequence_parallel(sp_size)
        sp_group = groups.get_sequence_parallel_group()
    _SP_GROUP[0] = sp_group
    from transformers.modeling_utils import AttentionInterface
    if _IMPL_NAME not in AttentionInterface._global_mapping:
        AttentionInterface.register(_IMPL_NAME, _ulysses_attention)
    kv = getattr(model.config, "num_key_value_heads", None) or \
        model.config.num_attention_heads
    import torch.distributed as dist
    world = dist.get_world_size(sp_group)
    assert kv % world == 0, \
        f"kv heads {kv} not divisible by sp degree {world}"
    model.set_attn_implementation(_IMPL_NAME) \
        if hasattr(model, "set_attn_implementation") else None
    model.config._attn_implementation = _IMPL_NAME
    log_dist(f"Ulysses-SP(HF): degree {world} on "
             f"{type(model).__name__}", ranks=[0])
    return model


def shard_batch_for_sp(batch, sp_group=None):
    """Shard input_ids/labels/position_ids along the sequence.

    Labels are causally pre-shifted (next-token) and the model must be
    given them with shift disabled — pass `shift_labels=False`-style
    loss, or compare logits externally. position_ids carry the GLOBAL
    positions of the local shard."""
    if sp_group is None:
        sp_group = groups.get_sequence_parallel_group()
    import torch.distributed as dist
    world = dist.get_world_size(sp_group)
    rank = dist.get_rank(sp_group)
    out = {}
    seq_len = batch["input_ids"].shape[1]
    assert seq_len % world == 0, f"seq {seq_len} % sp {world} != 0"
    s = seq_len // world
    sl = slice(rank * s, (rank + 1) * s)
    out["input_ids"] = batch["input_ids"][:, sl]
    out["position_ids"] = torch.arange(
        rank * s, (rank + 1) * s,
        device=batch["input_ids"].device).unsqueeze(0) \
        .expand(batch["input_ids"].shape[0], -1)
    if "labels" in batch:
        shifted = torch.roll(batch["labels"], shifts=-1, dims=1)
        shifted[:, -1] = -100
        out["shift_labels"] = shifted[:, sl]
    return out
