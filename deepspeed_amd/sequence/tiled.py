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
