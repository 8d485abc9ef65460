"""Llama-family causal LM, MI355X-native.

Built directly on the deepspeed_amd op set (HIP RMSNorm / RoPE / SwiGLU /
fused cross-entropy; hipBLASLt GEMMs via F.linear). Used by bench.py for the
BASELINE config "Llama-3 8B ZeRO-3 bf16".
"""
from dataclasses import dataclass, field

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.attention import flash_attention
from ..ops.functional import (apply_rope, build_rope_cache,
                              fused_cross_entropy, rms_norm, swiglu)


@dataclass
class LlamaConfig:
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_hidden_layers: int = 32
    num_attention_heads: int = 32
    num_key_value_heads: int = 8
    vocab_size: int = 128256
    max_position_embeddings: int = 8192
    rope_theta: float = 500000.0
    rms_norm_eps: float = 1e-5
    tie_word_embeddings: bool = False
    initializer_range: float = 0.02
    activation_checkpointing: bool = True
    attention_bias: bool = False  # Qwen2-style qkv bias

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads


LLAMA_CONFIGS = {
    "llama3-8b": LlamaConfig(),
    "llama3-70b": LlamaConfig(hidden_size=8192, intermediate_size=28672,
                              num_hidden_layers=80, num_attention_heads=64,
                              num_key_value_heads=8),
    "llama-tiny": LlamaConfig(hidden_size=256, intermediate_size=688,
                              num_hidden_layers=4, num_attention_heads=8,
                              num_key_value_heads=4, vocab_size=2048,
                              max_position_embeddings=512,
                              activation_checkpointing=False),
    "llama-small": LlamaConfig(hidden_size=1024, intermediate_size=2816,
                               num_hidden_layers=8, num_attention_heads=16,
                               num_key_value_heads=4, vocab_size=32000,
                               max_position_embeddings=4096),
    # Qwen2 family: llama architecture + qkv bias + tied tiny variants
    "qwen2-7b": LlamaConfig(hidden_size=3584, intermediate_size=18944,
                            num_hidden_layers=28, num_attention_heads=28,
                            num_key_value_heads=4, vocab_size=152064,
                            max_position_embeddings=32768,
                            rope_theta=1e6, rms_norm_eps=1e-6,
                            attention_bias=True),
    "qwen2-tiny": LlamaConfig(hidden_size=256, intermediate_size=688,
                              num_hidden_layers=4, num_attention_heads=8,
                              num_key_value_heads=4, vocab_size=2048,
                              max_position_embeddings=512,
                              attention_bias=True,
                              tie_word_embeddings=True,
                              activation_checkpointing=False),
}


class LlamaRMSNorm(nn.Module):
    def __init__(self, hidden_size, eps=1e-5):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(hidden_size))
        self.eps = eps

    def forward(self, x):
        return rms_norm(x, self.weight, self.eps)


class LlamaAttention(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        h, d = cfg.hidden_size, cfg.head_dim
        ab = getattr(cfg, "attention_bias", False)
        self.q_proj = nn.Linear(h, cfg.num_attention_heads * d, bias=ab)
        self.k_proj = nn.Linear(h, cfg.num_key_value_heads * d, bias=ab)
        self.v_proj = nn.Linear(h, cfg.num_key_value_heads * d, bias=ab)
        self.o_proj = nn.Linear(cfg.num_attention_heads * d, h, bias=False)
        self.sp_group = None  # set by enable_ulysses()
        self._dist_attn = None

    def forward(self, x, cos, sin, kv_cache=None):
        B, S, _ = x.shape
        d = self.cfg.head_dim
        q = self.q_proj(x).view(B, S, -1, d)
        k = self.k_proj(x).view(B, S, -1, d)
        v = self.v_proj(x).view(B, S, -1, d)
        # RoPE uses the caller's cos/sin slice (seq-offset aware under SP)
        q = apply_rope(q, cos, sin)
        k = apply_rope(k, cos, sin)
        if kv_cache is not None:
            k, v = kv_cache.update(k, v)
            rg = getattr(kv_cache, "ragged", None)
            if rg is not None:
                # ragged decode straight from the slot pool (no copies)
                from ..ops.loader import get_ext
                o = get_ext(required=True).ragged_decode(
                    q.contiguous(), *rg)
            else:
                mask = getattr(kv_cache, "attn_mask", None)
                causal = S > 1  # prefill is causal; decode sees all past
                o = flash_attention(q, k, v,
                                    causal=causal and mask is None,
                                    attn_mask=mask)
        elif self._dist_attn is not None:
            o = self._dist_attn(q, k, v, causal=True)
        else:
            o = flash_attention(q, k, v, causal=True)
        return self.o_proj(o.reshape(B, S, -1))


class StaticKVCache:
    """Graph-capturable cache: device-side position, static shapes.

    All updates are device-tensor indexed (index_copy_/index_fill_), so a
    decode step has no host-dependent control flow and can be captured in a
    hipGraph (torch.cuda.CUDAGraph on ROCm).
    """

    def __init__(self, batch, max_seq, n_kv, head_dim, dtype, device):
        self.k = torch.zeros(batch, max_seq, n_kv, head_dim, dtype=dtype,
                             device=device)
        self.v = torch.zeros_like(self.k)
        self.max_seq = max_seq

    def prefill(self, k, v):
        S = k.shape[1]
        self.k[:, :S] = k
        self.v[:, :S] = v

    def decode_update(self, k1, v1, pos_idx):
        """k1/v1 [B,1,Hk,D]; pos_idx int64 device scalar-tensor [1]."""
        self.k.index_copy_(1, pos_idx, k1)
        self.v.index_copy_(1, pos_idx, v1)
        return self.k, self.v


def _linear_1t(lin, x):
    """Decode-time linear: skinny-GEMV HIP kernel when applicable."""
    if x.is_cuda and x.dtype == torch.bfloat16:
        from ..ops.loader import get_ext
        ext = get_ext(required=False)
        if ext is not None:
            B = x.numel() // x.shape[-1]
            y = ext.gemv_bf16(lin.weight, x.reshape(B, -1).contiguous())
            return y.reshape(*x.shape[:-1], lin.weight.shape[0])
    return lin(x)


def llama_decode_step(model, ids, caches, pos_idx, attn_mask, cos_t, sin_t):
    """One static-shape decode step (graph-capturable).

    ids [B,1] int64; pos_idx [1] int64 device; attn_mask [1,1,1,max_seq]
    additive; cos_t/sin_t: full rope tables on device.
    """
    core = model.model
    x = core.embed_tokens(ids)
    cos = cos_t.index_select(0, pos_idx)
    sin = sin_t.index_select(0, pos_idx)
    for layer, cache in zip(core.layers, caches):
        attn = layer.self_attn
        h = layer.input_layernorm(x)
        B = h.shape[0]
        d = attn.cfg.head_dim
        q = _linear_1t(attn.q_proj, h).view(B, 1, -1, d)
        k = _linear_1t(attn.k_proj, h).view(B, 1, -1, d)
        v = _linear_1t(attn.v_proj, h).view(B, 1, -1, d)
        q = apply_rope(q, cos, sin)
        k = apply_rope(k, cos, sin)
        kf, vf = cache.decode_update(k, v, pos_idx)
        o = F.scaled_dot_product_attention(
            q.transpose(1, 2), kf.transpose(1, 2), vf.transpose(1, 2),
            attn_mask=attn_mask,
            enable_gqa=(kf.shape[2] != q.shape[2]))
        x = x + _linear_1t(attn.o_proj, o.transpose(1, 2).reshape(B, 1, -1))
        h2 = layer.post_attention_layernorm(x)
        mlp = layer.mlp
        x = x + _linear_1t(mlp.down_proj,
                           swiglu(_linear_1t(mlp.gate_proj, h2),
                                  _linear_1t(mlp.up_proj, h2)))
    x = core.norm(x)
    return _linear_1t(model.lm_head, x)


def enable_ulysses(model, sp_group=None):
    """Route every attention through Ulysses head-scatter all-to-all."""
    from ..sequence.layer import DistributedAttention
    for mod in model.modules():
        if isinstance(mod, LlamaAttention):
            mod.sp_group = sp_group
            mod._dist_attn = DistributedAttention(flash_attention, sp_group)
    return model


class LlamaMLP(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.gate_proj = nn.Linear(cfg.hidden_size, cfg.intermediate_size,
                                   bias=False)
        self.up_proj = nn.Linear(cfg.hidden_size, cfg.intermediate_size,
                                 bias=False)
        self.down_proj = nn.Linear(cfg.intermediate_size, cfg.hidden_size,
                                   bias=False)

    def forward(self, x):
        return self.down_proj(swiglu(self.gate_proj(x), self.up_proj(x)))


class LlamaDecoderLayer(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.input_layernorm = LlamaRMSNorm(cfg.hidden_size, cfg.rms_norm_eps)
        self.self_attn = LlamaAttention(cfg)
        self.post_attention_layernorm = LlamaRMSNorm(cfg.hidden_size,
                                                     cfg.rms_norm_eps)
        self.mlp = LlamaMLP(cfg)

    def forward(self, x, cos, sin, kv_cache=None):
        x = x + self.self_attn(self.input_layernorm(x), cos, sin,
                               kv_cache=kv_cache)
        x = x + self.mlp(self.post_attention_layernorm(x))
        return x


class LlamaModel(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(
            [LlamaDecoderLayer(cfg) for _ in range(cfg.num_hidden_layers)])
        self.norm = LlamaRMSNorm(cfg.hidden_size, cfg.rms_norm_eps)
        cos, sin = build_rope_cache(cfg.max_position_embeddings, cfg.head_dim,
                                    cfg.rope_theta)
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)

    def forward(self, input_ids, seq_offset=0, kv_caches=None,
                positions=None):
        x = self.embed_tokens(input_ids)
        S = input_ids.shape[1]
        if positions is not None:  # per-row positions (ragged decode)
            cos = self.rope_cos[positions]
            sin = self.rope_sin[positions]
        else:
            cos = self.rope_cos[seq_offset:seq_offset + S]
            sin = self.rope_sin[seq_offset:seq_offset + S]
        for i, layer in enumerate(self.layers):
            if self.cfg.activation_checkpointing and self.training:
                x = torch.utils.checkpoint.checkpoint(
                    layer, x, cos, sin, use_reentrant=False)
            else:
                x = layer(x, cos, sin,
                          kv_cache=kv_caches[i] if kv_caches else None)
        return self.norm(x)


class LlamaForCausalLM(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        self.model = LlamaModel(cfg)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        if cfg.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight
        self.apply(self._init_weights)

    def _init_weights(self, module):
        std = self.cfg.initializer_range
        if isinstance(module, nn.Linear):
            module.weight.data.normal_(0.0, std)
            if module.bias is not None:
                module.bias.data.zero_()
        elif isinstance(module, nn.Embedding):
            module.weight.data.normal_(0.0, std)

    def forward(self, input_ids, labels=None, seq_offset=0, kv_caches=None,
                positions=None):
        h = self.model(input_ids, seq_offset=seq_offset, kv_caches=kv_caches,
                       positions=positions)
        logits = self.lm_head(h)
        if labels is None:
            return logits
        # next-token prediction: shift
        return fused_cross_entropy(logits[:, :-1, :], labels[:, 1:])
