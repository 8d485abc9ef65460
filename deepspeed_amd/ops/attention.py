"""Attention dispatch.

Round-1 status: forward/backward run through torch SDPA on ROCm (MIOpen/CK
path) until the hand-written CDNA4 flash kernel (csrc/attention.hip) lands;
the HIP flash-attention is the flagged next kernel (guide §B recipe:
8-wave 32x32 swapped-QK^T online softmax, ~900 TF measured ladder).
"""
import torch
import torch.nn.functional as F


def flash_attention(q, k, v, causal=True):
    """q [B,S,Hq,D], k/v [B,S,Hkv,D] -> [B,S,Hq,D]. GQA-aware."""
    # SDPA wants [B,H,S,D]
    qt = q.transpose(1, 2)
    kt = k.transpose(1, 2)
    vt = v.transpose(1, 2)
    out = F.scaled_dot_product_attention(
        qt, kt, vt, is_causal=causal,
        enable_gqa=(k.shape[2] != q.shape[2]))
    return out.transpose(1, 2)
