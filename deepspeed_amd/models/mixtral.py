"""Mixtral-style MoE causal LM on the native op set + deepspeed_amd MoE.

BASELINE config #4: "Mixtral 8x7B DeepSpeed-MoE expert-parallel all-to-all
on 8xMI355X" — 8 experts, top-2 gating, EP over the fully-connected xGMI
node (a2a dispatch/combine).
"""
from dataclasses import dataclass

import torch
import torch.nn as nn

from ..moe.layer import MoE
from ..ops.functional import fused_cross_entropy
from .llama import (LlamaAttention, LlamaConfig, LlamaRMSNorm, LlamaMLP,
                    LLAMA_CONFIGS)
from ..ops.functional import build_rope_cache


@dataclass
class MixtralConfig(LlamaConfig):
    num_experts: int = 8
    top_k: int = 2
    ep_size: int = 1
    capacity_factor: float = 1.25
    aux_loss_coef: float = 0.01


MIXTRAL_CONFIGS = {
    # 8x7B geometry (Mistral-7B base: hidden 4096, inter 14336, 32 layers)
    "mixtral-8x7b": MixtralConfig(
        hidden_size=4096, intermediate_size=14336, num_hidden_layers=32,
        num_attention_heads=32, num_key_value_heads=8, vocab_size=32000,
        rope_theta=1e6, num_experts=8, top_k=2),
    "mixtral-tiny": MixtralConfig(
        hidden_size=256, intermediate_size=512, num_hidden_layers=2,
        num_attention_heads=8, num_key_value_heads=4, vocab_size=2048,
        max_position_embeddings=512, num_experts=4, top_k=2,
        activation_checkpointing=False),
}


class MixtralDecoderLayer(nn.Module):
    def __init__(self, cfg: MixtralConfig):
        super().__init__()
        self.input_layernorm = LlamaRMSNorm(cfg.hidden_size, cfg.rms_norm_eps)
        self.self_attn = LlamaAttention(cfg)
        self.post_attention_layernorm = LlamaRMSNorm(cfg.hidden_size,
                                                     cfg.rms_norm_eps)
        expert = LlamaMLP(cfg)
        self.block_sparse_moe = MoE(cfg.hidden_size, expert,
                                    num_experts=cfg.num_experts,
                                    ep_size=cfg.ep_size, k=cfg.top_k,
                                    capacity_factor=cfg.capacity_factor)

    def forward(self, x, cos, sin, kv_cache=None):
        x = x + self.self_attn(self.input_layernorm(x), cos, sin,
                               kv_cache=kv_cache)
        moe_out, l_aux, _ = self.block_sparse_moe(
            self.post_attention_layernorm(x))
        return x + moe_out, l_aux


class MixtralForCausalLM(nn.Module):
    def __init__(self, cfg: MixtralConfig):
        super().__init__()
        self.cfg = cfg
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(
            [MixtralDecoderLayer(cfg) for _ in range(cfg.num_hidden_layers)])
        self.norm = LlamaRMSNorm(cfg.hidden_size, cfg.rms_norm_eps)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        cos, sin = build_rope_cache(cfg.max_position_embeddings, cfg.head_dim,
                                    cfg.rope_theta)
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)
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
        x = self.embed_tokens(input_ids)
        S = input_ids.shape[1]
        if positions is not None:  # per-row positions (ragged decode)
            cos = self.rope_cos[positions]
            sin = self.rope_sin[positions]
        else:
            cos = self.rope_cos[seq_offset:seq_offset + S]
            sin = self.rope_sin[seq_offset:seq_offset + S]
        aux_total = 0.0
        for i, layer in enumerate(self.layers):
            kv = kv_caches[i] if kv_caches is not None else None
            if self.cfg.activation_checkpointing and self.training \
                    and kv is None:
                x, l_aux = torch.utils.checkpoint.checkpoint(
                    layer, x, cos, sin, use_reentrant=False)
            else:
                x, l_aux = layer(x, cos, sin, kv_cache=kv)
            aux_total = aux_total + l_aux
        x = self.norm(x)
        logits = self.lm_head(x)
        if labels is None:
            return logits
        loss = fused_cross_entropy(logits[:, :-1, :], labels[:, 1:])
        return loss + self.cfg.aux_loss_coef * aux_total.to(loss.dtype) \
            / len(self.layers)
