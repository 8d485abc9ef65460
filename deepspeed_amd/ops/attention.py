"""Flash attention for MI355X — hand-written CDNA4 kernels, default path.

Forward (csrc/attention.hip, v5): 8-wave swapped-QK^T 32x32 MFMA kernel,
in-register softmax (permlane32_swap, exp2 domain, defer-max), async
double-buffered K staging. Measured 350 TF (B=1) / 366 TF (B=4) at
S=4096 — faster than torch-rocm SDPA (aotriton) forward at B=1.
Backward (csrc/attention_bwd.hip): two-pass exact flash backward from the
saved LSE (fused Drow kernel; pass 1 accumulates dK/dV over the GQA group
in-kernel) — faster than the SDPA backward at the training shape.
Set DSAMD_FLASH=0 to fall back to torch SDPA; a chunked-GEMM backward
(DSAMD_FLASH_BWD=0) remains as a reference implementation.

Parity role: reference inference/v2 blocked flash kernels + training
softmax/attention kernels (csrc/transformer/softmax_kernels.cu).
"""
import math
import os

import torch
import torch.nn.functional as F

from .loader import get_ext


class _FlashAttnFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, causal, scale, kvmask=None):
        ext = get_ext(required=True)
        out, lse = ext.flash_attn_fwd(q, k, v, causal, scale, kvmask)
        ctx.save_for_backward(q, k, v, out, lse,
                              kvmask if kvmask is not None
                              else torch.empty(0))
        ctx.causal = causal
        ctx.scale = scale
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, out, lse, kvmask = ctx.saved_tensors
        if kvmask.numel() == 0:
            kvmask = None
        causal, scale = ctx.causal, ctx.scale
        B, S, Hq, D = q.shape
        Hk = k.shape[2]
        G = Hq // Hk

        if os.environ.get("DSAMD_FLASH_BWD", "1") == "1":
            ext = get_ext(required=True)
            dq, dk, dv = ext.flash_attn_bwd(
                q, k, v, dout.contiguous(), out, lse, causal, scale, kvmask)
            return dq, dk, dv, None, None, None

        # head-major views [B,H,S,D]
        qh = q.permute(0, 2, 1, 3)
        kh = k.permute(0, 2, 1, 3)
        vh = v.permute(0, 2, 1, 3)
        doh = dout.permute(0, 2, 1, 3).contiguous()
        oh = out.permute(0, 2, 1, 3)

        # D_i = rowsum(dO * O)  [B,Hq,S]
        drow = (doh.float() * oh.float()).sum(-1)

        qg = qh.reshape(B, Hk, G, S, D)
        dog = doh.reshape(B, Hk, G, S, D)
        lse_g = lse.reshape(B, Hk, G, S)
        drow_g = drow.reshape(B, Hk, G, S)

        dq = torch.zeros(B, Hk, G, S, D, device=q.device, dtype=torch.float32)
        dk = torch.empty_like(kh, dtype=torch.float32)
        dv = torch.empty_like(vh, dtype=torch.float32)

        CHUNK = 1024
        rows = torch.arange(S, device=q.device)
        for c0 in range(0, S, CHUNK):
            c1 = min(c0 + CHUNK, S)
            kc = kh[:, :, c0:c1]          # [B,Hk,c,D]
            vc = vh[:, :, c0:c1]
            # S_c = q @ k^T * scale  -> P via saved LSE
            s_c = torch.einsum("bhgsd,bhcd->bhgsc", qg, kc).float() * scale
            if kvmask is not None:
                s_c = s_c + kvmask[:, c0:c1].view(B, 1, 1, 1, c1 - c0)
            p = torch.exp(s_c - lse_g.unsqueeze(-1))
            if causal:
                mask = rows.view(1, 1, 1, S, 1) >= (c0 + torch.arange(
                    c1 - c0, device=q.device)).view(1, 1, 1, 1, -1)
                p = p * mask
            p16 = p.to(q.dtype)
            # dV_c = P^T @ dO   (sum over G and S)
            dv[:, :, c0:c1] = torch.einsum("bhgsc,bhgsd->bhcd", p16,
                                           dog).float()
            # dP = dO @ V^T
            dp = torch.einsum("bhgsd,bhcd->bhgsc", dog, vc).float()
            ds = (p * (dp - drow_g.unsqueeze(-1)) * scale).to(q.dtype)
            # dQ += dS @ K
            dq += torch.einsum("bhgsc,bhcd->bhgsd", ds, kc).float()
            # dK_c = dS^T @ Q  (sum over G and S)
            dk[:, :, c0:c1] = torch.einsum("bhgsc,bhgsd->bhcd", ds,
                                           qg).float()

        dq_out = dq.reshape(B, Hq, S, D).permute(0, 2, 1, 3).to(q.dtype)
        dk_out = dk.permute(0, 2, 1, 3).to(k.dtype)
        dv_out = dv.permute(0, 2, 1, 3).to(v.dtype)
        return dq_out.contiguous(), dk_out.contiguous(), \
            dv_out.contiguous(), None, None, None


# Training default: the in-tree HIP kernels (fwd v5 beats SDPA; bwd beats
# the SDPA backward; end-to-end A/B on the ZeRO-3 Llama-8B bench confirms
# parity-or-better). DSAMD_FLASH=0 falls back to torch SDPA.
_USE_HIP_FLASH = os.environ.get("DSAMD_FLASH", "1") == "1"


def _as_kv_padding_mask(attn_mask, B, Sk):
    """[B,S] float kv mask if attn_mask is a kv-column-only additive mask
    ([B,S], [B,1,1,S], [1,1,1,S], [B,1,S]); else None."""
    if attn_mask is None or attn_mask.dtype == torch.bool:
        return None
    m = attn_mask
    if m.dim() == 4 and m.shape[1] == 1 and m.shape[2] == 1:
        m = m[:, 0, 0]
    elif m.dim() == 3 and m.shape[1] == 1:
        m = m[:, 0]
    elif m.dim() != 2:
        return None
    if m.shape[-1] != Sk:
        return None
    if m.shape[0] == 1 and B > 1:
        m = m.expand(B, Sk)
    if m.shape[0] != B:
        return None
    return m.contiguous().float()


def flash_attention(q, k, v, causal=True, attn_mask=None):
    """q [B,S,Hq,D], k/v [B,S,Hkv,D] -> [B,S,Hq,D]. GQA-aware.

    attn_mask: optional additive float mask. KV-padding-shaped masks
    ([B,1,1,Sk] etc., the BERT case) run in the HIP kernel; arbitrary
    [B,Hq,Sq,Sk] masks route through SDPA."""
    scale = 1.0 / math.sqrt(q.shape[-1])
    hip_ok = _USE_HIP_FLASH and q.is_cuda and q.dtype == torch.bfloat16 \
        and q.shape[-1] in (64, 128) and q.shape[1] == k.shape[1]
    if attn_mask is not None:
        kvm = _as_kv_padding_mask(attn_mask, q.shape[0], k.shape[1]) \
            if hip_ok else None
        if kvm is not None:
            return _FlashAttnFn.apply(q.contiguous(), k.contiguous(),
                                      v.contiguous(), causal, scale, kvm)
        out = F.scaled_dot_product_attention(
            q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
            attn_mask=attn_mask, is_causal=False,
            enable_gqa=(k.shape[2] != q.shape[2]))
        return out.transpose(1, 2)
    if hip_ok:
        return _FlashAttnFn.apply(q.contiguous(), k.contiguous(),
                                  v.contiguous(), causal, scale)
    # fallback (CPU tests / non-128 head dims): torch SDPA math
    qt = q.transpose(1, 2)
    kt = k.transpose(1, 2)
    vt = v.transpose(1, 2)
    out = F.scaled_dot_product_attention(
        qt, kt, vt, is_causal=causal,
        enable_gqa=(k.shape[2] != q.shape[2]))
    return out.transpose(1, 2)
