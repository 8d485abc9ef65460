"""DS4Sci_EvoformerAttention — biased multi-head attention for Evoformer
stacks (OpenFold/AlphaFold-style triangle + MSA attention).

Parity: reference `deepspeed/ops/deepspeed4science/evoformer_attn.py`
(DS4Sci_EvoformerAttention, csrc/deepspeed4science/evoformer_attn —
CUTLASS fMHA with additive biases). Semantics: softmax(QK^T/sqrt(d)
+ bias_1 + bias_2) V, where each bias broadcasts against the [.., H, Q, K]
logits; bias_2 (the pair representation bias) receives gradients, bias_1
(the mask bias) does not.

MI355X note: this runs through torch-rocm matmul/softmax (MFMA-backed
hipBLASLt GEMMs). Evoformer shapes are many small heads (D=8..32), which
the 128-head-dim flash kernel does not cover; a dedicated CDNA4 kernel is
a ROADMAP item. Numerics match the reference op: fp32 softmax
accumulation regardless of input dtype.
"""
import math

import torch


def DS4Sci_EvoformerAttention(Q, K, V, biases):
    """Q,K,V: [*, Q/K, H, D]; biases: list of tensors broadcastable to
    [*, H, Q, K] (use None entries to skip). Returns [*, Q, H, D]."""
    assert Q.dim() >= 3
    d = Q.shape[-1]
    scale = 1.0 / math.sqrt(d)
    # [*, H, Q, D] layout for the GEMMs
    q = Q.transpose(-2, -3)
    k = K.transpose(-2, -3)
    v = V.transpose(-2, -3)
    logits = torch.matmul(q.float() * scale, k.float().transpose(-1, -2))
    for b in biases:
        if b is not None:
            logits = logits + b.float()
    probs = torch.softmax(logits, dim=-1)
    out = torch.matmul(probs.to(v.dtype), v)
    return out.transpose(-2, -3).contiguous()
