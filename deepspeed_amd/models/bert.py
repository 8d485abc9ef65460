"""BERT-family encoder on the native MI355X op set.

Role parity: the reference's training transformer kernel targets exactly
this architecture (csrc/transformer/ fused BERT encoder layer,
DeepSpeedTransformerLayer, ops/transformer/transformer.py:296) and the
BERT rows in BASELINE.md (pretrain wall-clock, SQuAD fine-tune).

MI355X-native: attention runs the in-tree flash kernel at head_dim 64
with the [B,S] additive kv padding mask IN-KERNEL (attention.hip); norms
are the fused HIP layer_norm; GELU stays an elementwise torch op (fused
by inductor under torch.compile).
"""
from dataclasses import dataclass

import torch
from torch import nn

from ..ops.attention import flash_attention
from ..ops.functional import layer_norm


@dataclass
class BertConfig:
    vocab_size: int = 30522
    hidden_size: int = 1024
    num_hidden_layers: int = 24
    num_attention_heads: int = 16
    intermediate_size: int = 4096
    max_position_embeddings: int = 512
    type_vocab_size: int = 2
    layer_norm_eps: float = 1e-12
    initializer_range: float = 0.02

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads


BERT_CONFIGS = {
    "bert-large": BertConfig(),
    "bert-base": BertConfig(hidden_size=768, num_hidden_layers=12,
                            num_attention_heads=12,
                            intermediate_size=3072),
    "bert-tiny": BertConfig(vocab_size=1024, hidden_size=128,
                            num_hidden_layers=2, num_attention_heads=2,
                            intermediate_size=256,
                            max_position_embeddings=128),
}


class BertLayerNorm(nn.Module):
    def __init__(self, hidden, eps):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(hidden))
        self.bias = nn.Parameter(torch.zeros(hidden))
        self.eps = eps

    def forward(self, x):
        return layer_norm(x, self.weight, self.bias, self.eps)


class BertSelfAttention(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        h = cfg.hidden_size
        self.qkv = nn.Linear(h, 3 * h)
        self.out = nn.Linear(h, h)
        self.n_heads = cfg.num_attention_heads
        self.head_dim = cfg.head_dim

    def forward(self, x, kv_mask=None):
        B, S, H = x.shape
        q, k, v = self.qkv(x).chunk(3, dim=-1)
        q = q.view(B, S, self.n_heads, self.head_dim)
        k = k.view(B, S, self.n_heads, self.head_dim)
        v = v.view(B, S, self.n_heads, self.head_dim)
        # bidirectional attention; [B,S] additive padding mask runs
        # inside the flash kernel (attention.hip kvmask path)
        o = flash_attention(q, k, v, causal=False,
                            attn_mask=None if kv_mask is None
                            else kv_mask.view(B, 1, 1, S))
        return self.out(o.reshape(B, S, H))


class BertLayer(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.attention = BertSelfAttention(cfg)
        self.attn_norm = BertLayerNorm(cfg.hidden_size, cfg.layer_norm_eps)
        self.intermediate = nn.Linear(cfg.hidden_size,
                                      cfg.intermediate_size)
        self.output = nn.Linear(cfg.intermediate_size, cfg.hidden_size)
        self.out_norm = BertLayerNorm(cfg.hidden_size, cfg.layer_norm_eps)

    def forward(self, x, kv_mask=None):
        x = self.attn_norm(x + self.attention(x, kv_mask))
        h = self.output(torch.nn.functional.gelu(self.intermediate(x)))
        return self.out_norm(x + h)


class BertModel(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.cfg = cfg
        self.word_embeddings = nn.Embedding(cfg.vocab_size,
                                            cfg.hidden_size)
        self.position_embeddings = nn.Embedding(
            cfg.max_position_embeddings, cfg.hidden_size)
        self.token_type_embeddings = nn.Embedding(cfg.type_vocab_size,
                                                  cfg.hidden_size)
        self.embed_norm = BertLayerNorm(cfg.hidden_size,
                                        cfg.layer_norm_eps)
        self.layers = nn.ModuleList(
            [BertLayer(cfg) for _ in range(cfg.num_hidden_layers)])

    def forward(self, input_ids, attention_mask=None, token_type_ids=None):
        B, S = input_ids.shape
        pos = torch.arange(S, device=input_ids.device)
        x = self.word_embeddings(input_ids) + \
            self.position_embeddings(pos)[None]
        if token_type_ids is not None:
            x = x + self.token_type_embeddings(token_type_ids)
        x = self.embed_norm(x)
        kv_mask = None
        if attention_mask is not None:
            # HF-style 1/0 mask -> additive float [B,S]
            kv_mask = torch.where(attention_mask.bool(), 0.0,
                                  float("-inf")).to(torch.float32)
        for layer in self.layers:
            x = layer(x, kv_mask)
        return x


class BertForPreTraining(nn.Module):
    """MLM head (tied decoder) over the encoder."""

    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.cfg = cfg
        self.bert = BertModel(cfg)
        self.transform = nn.Linear(cfg.hidden_size, cfg.hidden_size)
        self.transform_norm = BertLayerNorm(cfg.hidden_size,
                                            cfg.layer_norm_eps)
        self.decoder = nn.Linear(cfg.hidden_size, cfg.vocab_size)
        self.decoder.weight = self.bert.word_embeddings.weight
        self.apply(self._init)

    def _init(self, m):
        if isinstance(m, (nn.Linear, nn.Embedding)):
            m.weight.data.normal_(0.0, self.cfg.initializer_range)
            if isinstance(m, nn.Linear) and m.bias is not None:
                m.bias.data.zero_()

    def forward(self, input_ids, attention_mask=None, token_type_ids=None,
                labels=None):
        h = self.bert(input_ids, attention_mask, token_type_ids)
        logits = self.decoder(torch.nn.functional.gelu(
            self.transform_norm(self.transform(h))))
        if labels is None:
            return logits
        return torch.nn.functional.cross_entropy(
            logits.float().view(-1, self.cfg.vocab_size),
            labels.reshape(-1), ignore_index=-100)


class BertForQuestionAnswering(nn.Module):
    """SQuAD-style span head (BASELINE BingBertSquad row)."""

    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.cfg = cfg
        self.bert = BertModel(cfg)
        self.qa_outputs = nn.Linear(cfg.hidden_size, 2)

    def forward(self, input_ids, attention_mask=None, token_type_ids=None,
                start_positions=None, end_positions=None):
        h = self.bert(input_ids, attention_mask, token_type_ids)
        start, end = self.qa_outputs(h).split(1, dim=-1)
        start, end = start.squeeze(-1), end.squeeze(-1)
        if start_positions is None:
            return start, end
        loss = (torch.nn.functional.cross_entropy(start.float(),
                                                  start_positions) +
                torch.nn.functional.cross_entropy(end.float(),
                                                  end_positions)) / 2
        return loss
