"""FPDT-style chunked attention with host offload of KV chunks.

Parity: reference `sequence/fpdt_layer.py` (`update_out_and_lse:59`,
`_FPDTGPUOffloadingAttentionImpl_:545`, `SequenceChunk:497`). Processes
multi-million-token sequences by iterating KV in chunks that live in pinned
host memory; partial attention outputs merge with the online-softmax
(out, lse) update. Works on top of Ulysses groups (per-rank local compute).
"""
import math

import torch

from ..ops.attention import flash_attention


def update_out_and_lse(out, lse, new_out, new_lse):
    """Merge partial attention results (log-sum-exp weighted).

    out [B,S,H,D] fp32, lse [B,H,S] fp32.
    """
    new_out = new_out.float()
    if out is None:
        return new_out, new_lse
    merged_lse = torch.logaddexp(lse, new_lse)
    w_old = torch.exp(lse - merged_lse).permute(0, 2, 1).unsqueeze(-1)
    w_new = torch.exp(new_lse - merged_lse).permute(0, 2, 1).unsqueeze(-1)
    return out * w_old + new_out * w_new, merged_lse


def _sdpa_with_lse(q, k, v, causal):
    """Partial attention + per-row LSE via fp32 math (chunk-sized inputs)."""
    qt = q.permute(0, 2, 1, 3).float()
    kt = k.permute(0, 2, 1, 3).float()
    vt = v.permute(0, 2, 1, 3).float()
    G = qt.shape[1] // kt.shape[1]
    if G > 1:
        kt = kt.repeat_interleave(G, dim=1)
        vt = vt.repeat_interleave(G, dim=1)
    s = qt @ kt.transpose(-1, -2) / math.sqrt(q.shape[-1])
    if causal is not None:
        s = s + causal  # additive mask (broadcast [Sq, Skv])
    lse = torch.logsumexp(s, dim=-1)          # [B,H,Sq]
    p = torch.exp(s - lse.unsqueeze(-1))
    o = (p @ vt).permute(0, 2, 1, 3)          # [B,Sq,H,D]
    return o, lse


class SequenceChunk:
    """A KV chunk parked in pinned host memory (ref SequenceChunk:497)."""

    def __init__(self, k, v):
        pin = k.is_cuda
        self.k_cpu = k.detach().to("cpu", non_blocking=False).pin_memory() \
            if pin else k.detach().cpu()
        self.v_cpu = v.detach().to("cpu", non_blocking=False).pin_memory() \
            if pin else v.detach().cpu()
        self.device = k.device
        self.dtype = k.dtype

    def fetch(self):
        return (self.k_cpu.to(self.device, non_blocking=True),
                self.v_cpu.to(self.device, non_blocking=True))


@torch.no_grad()
def fpdt_attention(q, k, v, chunk_size=4096, causal=True,
                   offload_to_host=False):
    """Chunked causal attention: O(chunk) device memory for scores.

    q,k,v [B,S,H,D]; iterates q chunks x kv chunks with online-softmax
    merging; optionally parks KV chunks in host memory between uses.
    """
    B, S, Hq, D = q.shape
    n_chunks = (S + chunk_size - 1) // chunk_size
    if offload_to_host:
        kv_chunks = [SequenceChunk(k[:, i * chunk_size:(i + 1) * chunk_size],
                                   v[:, i * chunk_size:(i + 1) * chunk_size])
                     for i in range(n_chunks)]
    outs = []
    for qi in range(n_chunks):
        q0, q1 = qi * chunk_size, min((qi + 1) * chunk_size, S)
        qc = q[:, q0:q1]
        out, lse = None, None
        for ki in range(qi + 1 if causal else n_chunks):
            k0, k1 = ki * chunk_size, min((ki + 1) * chunk_size, S)
            if offload_to_host:
                kc, vc = kv_chunks[ki].fetch()
            else:
                kc, vc = k[:, k0:k1], v[:, k0:k1]
            if causal and ki == qi:
                rows = torch.arange(q0, q1, device=q.device)
                cols = torch.arange(k0, k1, device=q.device)
                mask = torch.where(rows.view(-1, 1) >= cols.view(1, -1),
                                   0.0, float("-inf"))
            elif causal and ki > qi:
                continue
            else:
                mask = None
            po, plse = _sdpa_with_lse(qc, kc, vc, mask)
            out, lse = update_out_and_lse(out, lse, po, plse)
        outs.append(out.to(q.dtype))
    return torch.cat(outs, dim=1)


# ---- chunked FFN + logits-loss (ref fpdt_layer.py FPDT_FFN:1127,
# FPDT_LogitsLoss:1208) — reuse the tiled-compute machinery: shards are
# checkpointed segments of one autograd graph (recompute-in-backward).
def fpdt_ffn(ffn_module, x, chunk_size=4096):
    """Run an FFN over the sequence in chunks (activation memory bounded
    by one chunk)."""
    import math

    from .tiled import sequence_tiled_compute
    shards = max(1, math.ceil(x.shape[1] / chunk_size))
    return sequence_tiled_compute(ffn_module, x, shards, dim=1,
                                  compute_params=True)


def fpdt_logits_loss(hidden, lm_weight, labels, chunk_size=4096,
                     shift_labels=True, ignore_index=-100):
    """CE loss over chunked logits — the [B, S, V] tensor never
    materializes at once."""
    import math

    from .tiled import tiled_logits_loss
    shards = max(1, math.ceil(hidden.shape[1] / chunk_size))
    return tiled_logits_loss(hidden, lm_weight, labels,
                             num_shards=shards, shift_labels=shift_labels,
                             ignore_index=ignore_index)
