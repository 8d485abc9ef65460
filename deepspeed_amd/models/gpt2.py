"""GPT-2 family on the native op set.

BASELINE.json's plumbing config ("GPT-2 small ZeRO-1 on gloo
world_size=2") and a second from-scratch model family exercising the
LayerNorm + GELU + learned-positional path of the kernel set (RMSNorm/
RoPE/SwiGLU are covered by the Llama family).
"""
from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.functional import fused_cross_entropy, layer_norm


@dataclass
class GPT2Config:
    hidden_size: int = 768
    num_hidden_layers: int = 12
    num_attention_heads: int = 12
    vocab_size: int = 50257
    max_position_embeddings: int = 1024
    layer_norm_eps: float = 1e-5
    initializer_range: float = 0.02
    activation_checkpointing: bool = False

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads


GPT2_CONFIGS = {
    "gpt2-small": GPT2Config(),
    "gpt2-medium": GPT2Config(hidden_size=1024, num_hidden_layers=24,
                              num_attention_heads=16),
    "gpt2-tiny": GPT2Config(hidden_size=128, num_hidden_layers=2,
                            num_attention_heads=4, vocab_size=512,
                            max_position_embeddings=128),
}


class GPT2Attention(nn.Module):
    def __init__(self, cfg):
        super().__init__()
        self.cfg = cfg
        self.c_attn = nn.Linear(cfg.hidden_size, 3 * cfg.hidden_size)
        self.c_proj = nn.Linear(cfg.hidden_size, cfg.hidden_size)

    def forward(self, x):
        B, S, H = x.shape
        qkv = self.c_attn(x)
        q, k, v = qkv.split(H, dim=-1)
        nh, d = self.cfg.num_attention_heads, self.cfg.head_dim
        q = q.view(B, S, nh, d).transpose(1, 2)
        k = k.view(B, S, nh, d).transpose(1, 2)
        v = v.view(B, S, nh, d).transpose(1, 2)
        o = F.scaled_dot_product_attention(q, k, v, is_causal=True)
        return self.c_proj(o.transpose(1, 2).reshape(B, S, H))


class GPT2MLP(nn.Module):
    def __init__(self, cfg):
        super().__init__()
        self.c_fc = nn.Linear(cfg.hidden_size, 4 * cfg.hidden_size)
        self.c_proj = nn.Linear(4 * cfg.hidden_size, cfg.hidden_size)

    def forward(self, x):
        return self.c_proj(F.gelu(self.c_fc(x), approximate="tanh"))


class GPT2Block(nn.Module):
    def __init__(self, cfg):
        super().__init__()
        self.ln_1 = nn.LayerNorm(cfg.hidden_size, eps=cfg.layer_norm_eps)
        self.attn = GPT2Attention(cfg)
        self.ln_2 = nn.LayerNorm(cfg.hidden_size, eps=cfg.layer_norm_eps)
        self.mlp = GPT2MLP(cfg)

    def forward(self, x):
        x = x + self.attn(layer_norm(x, self.ln_1.weight, self.ln_1.bias,
                                     self.ln_1.eps))
        x = x + self.mlp(layer_norm(x, self.ln_2.weight, self.ln_2.bias,
                                    self.ln_2.eps))
        return x


class GPT2LMHeadModel(nn.Module):
    def __init__(self, cfg: GPT2Config):
        super().__init__()
        self.cfg = cfg
        self.wte = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.wpe = nn.Embedding(cfg.max_position_embeddings, cfg.hidden_size)
        self.h = nn.ModuleList(
            [GPT2Block(cfg) for _ in range(cfg.num_hidden_layers)])
        self.ln_f = nn.LayerNorm(cfg.hidden_size, eps=cfg.layer_norm_eps)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        self.lm_head.weight = self.wte.weight  # GPT-2 ties embeddings
        self.apply(self._init)

    def _init(self, m):
        if isinstance(m, (nn.Linear, nn.Embedding)):
            m.weight.data.normal_(0.0, self.cfg.initializer_range)
        if isinstance(m, nn.Linear) and m.bias is not None:
            m.bias.data.zero_()

    def forward(self, input_ids, labels=None):
        S = input_ids.shape[1]
        pos = torch.arange(S, device=input_ids.device)
        x = self.wte(input_ids) + self.wpe(pos)
        for blk in self.h:
            if self.cfg.activation_checkpointing and self.training:
                x = torch.utils.checkpoint.checkpoint(blk, x,
                                                      use_reentrant=False)
            else:
                x = blk(x)
        x = layer_norm(x, self.ln_f.weight, self.ln_f.bias, self.ln_f.eps)
        logits = self.lm_head(x)
        if labels is None:
            return logits
        return fused_cross_entropy(logits[:, :-1, :], labels[:, 1:])
