"""Tiled compute for long sequences (Ulysses-SP / ALST memory tools).

Parity: reference `runtime/sequence_parallel/ulysses_sp.py:720`
(sequence_tiled_compute), `:943` (TiledMLP), `:1065` (TiledFusedLogitsLoss).

Bounds activation memory by processing the sequence in shards with
recompute-in-backward (checkpoint) per shard — e.g. 32k-token logits
(32k x 128k bf16 = 8.4 GB) never materialize at once.
"""
import torch

from ..ops.functional import fused_cross_entropy


def sequence_tiled_compute(fn, x, num_shards, dim=1, compute_params=None):
    """Apply fn shard-by-shard along `dim` with per-shard recompute."""
    chunks = x.tensor_split(num_shards, dim=dim)
    outs = []
    for c in chunks:
        if torch.is_grad_enabled() and (c.requires_grad or
                                        compute_params is not None):
            outs.append(torch.utils.checkpoint.checkpoint(
                fn, c, use_reentrant=False))
        else:
            outs.append(fn(c))
    return torch.cat(outs, dim=dim)


class TiledMLP(torch.nn.Module):
    """Wrap an MLP module so its forward runs in sequence shards."""

    def __init__(self, mlp, num_shards=4):
        super().__init__()
        self.mlp = mlp
        self.num_shards = num_shards

    def forward(self, x):
        return sequence_tiled_compute(self.mlp, x, self.num_shards,
                                      dim=1, compute_params=True)


def tiled_logits_loss(hidden, lm_weight, labels, num_shards=8,
                      shift_labels=True, ignore_index=-100):
    """CE over tiled logits: hidden [B,S,H] @ lm_weight^T -> loss, without
    materializing [B,S,V] at once.

    Returns the mean loss over valid (non-ignored) tokens.
    """
    B, S, H = hidden.shape
    if shift_labels:
        hidden = hidden[:, :-1, :]
        labels = labels[:, 1:]
        S = S - 1
    h_chunks = hidden.tensor_split(num_shards, dim=1)
    l_chunks = labels.tensor_split(num_shards, dim=1)
    total_valid = (labels != ignore_index).sum().clamp(min=1)

    def chunk_loss(h, lab, n_valid_chunk):
        logits = torch.nn.functional.linear(h, lm_weight)
        # sum-loss over chunk (scaled later by global valid count)
        loss_mean = fused_cross_entropy(logits, lab,
                                        ignore_index=ignore_index)
        return loss_mean * n_valid_chunk

    total = hidden.new_zeros((), dtype=torch.float32)
    for h, lab in zip(h_chunks, l_chunks):
        nv = (lab != ignore_index).sum().clamp(min=1)
        if torch.is_grad_enabled() and (h.requires_grad or
                                        lm_weight.requires_grad):
            part = torch.utils.checkpoint.checkpoint(
                chunk_loss, h, lab, nv, use_reentrant=False)
        else:
            part = chunk_loss(h, lab, nv)
        total = total + part.float()
    return (total / total_valid).to(hidden.dtype)


def agree_num_shards(num_shards, group=None):
    """All SP ranks must tile identically or collectives inside fn
    deadlock (ref ulysses_sp.py TiledMLP example): take the group MAX."""
    import torch.distributed as dist
    if not dist.is_initialized() or dist.get_world_size(group) == 1:
        return num_shards
    t = torch.tensor(int(num_shards))
    if torch.cuda.is_available():
        t = t.cuda()
    dist.all_reduce(t, op=dist.ReduceOp.MAX, group=group)
    return int(t.item())


def enable_tiled_mlp_for_hf(model, num_shards=None, sp_group=None):
    """Monkey-patch gate/up/down MLP modules of an HF model to run
    sequence-tiled (ref ulysses_sp.py:943 TiledMLP HF example).

    Design note vs the reference: the reference re-runs backward per
    shard via a custom autograd.Function and gates ZeRO's per-param
    reduction with ds_grad_is_ready; here each shard is a checkpointed
    segment of ONE autograd graph, so grads accumulate exactly once and
    the ZeRO hooks need no gating."""
    import math
    n_patched = 0
    for mod in model.modules():
        if all(hasattr(mod, a) for a in ("gate_proj", "up_proj",
                                         "down_proj")):
            inner = mod.forward

            def tiled_forward(x, _inner=inner, _shards=num_shards,
                              _spg=sp_group):
                s = _shards or max(1, math.ceil(x.shape[-2] /
                                                max(x.shape[-1], 1)))
                s = agree_num_shards(s, _spg)
                return sequence_tiled_compute(_inner, x, s, dim=-2,
                                              compute_params=True)

            mod.forward = tiled_forward
            n_patched += 1
    return n_patched
