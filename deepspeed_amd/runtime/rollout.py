"""Rollout engine interface for RLHF trainers.

Parity: reference `runtime/rollout/base.py` (RolloutConfig /
SamplingConfig / RolloutRequest / RolloutBatch / RolloutEngine) and
`runtime/rollout/hybrid_engine_rollout.py`. The trainer loop talks to
generation through these small dataclasses so engine-specific concerns
(sharding, KV caches, graphs) stay out of it.

MI355X departure: the co-located backend generates with the ZeRO-3
training shards gathered IN PLACE (runtime/hybrid_engine.py) — there is
no separate inference copy to `sync_weights` into, so weight sync is
free by construction. Instead of attention-masked left padding, prompts
are bucketed by true length and each bucket runs unpadded (pad tokens
never enter the KV cache); the returned batch is right-padded with
`response_start_idx` giving each row's true prompt length.
"""
from abc import ABC, abstractmethod
from dataclasses import dataclass

import torch


@dataclass
class RolloutConfig:
    engine: str = "hybrid_engine"
    use_graph_capture: bool = False  # decode hipGraph capture


@dataclass
class SamplingConfig:
    max_new_tokens: int
    temperature: float = 1.0
    top_p: float = 1.0
    top_k: int = -1
    n_samples_per_prompt: int = 1


@dataclass
class RolloutRequest:
    """Prompts, left-padded with `pad_token_id` (real tokens at the
    right edge) as RLHF data loaders conventionally emit them."""
    prompt_ids: torch.Tensor            # [B, T_p]
    prompt_attention_mask: torch.Tensor  # [B, T_p], 1 on real tokens

    def __post_init__(self):
        if self.prompt_ids.dim() != 2:
            raise ValueError(f"prompt_ids must be [B, T_p]; got "
                             f"{tuple(self.prompt_ids.shape)}")
        if self.prompt_attention_mask.shape != self.prompt_ids.shape:
            raise ValueError("prompt_attention_mask shape "
                             f"{tuple(self.prompt_attention_mask.shape)}"
                             f" != {tuple(self.prompt_ids.shape)}")


@dataclass
class RolloutBatch:
    """Prompt+response per row, right-padded to the longest sequence.
    B' = B * n_samples_per_prompt; samples of one prompt are adjacent."""
    input_ids: torch.Tensor        # [B', T]
    attention_mask: torch.Tensor   # [B', T]
    response_start_idx: torch.Tensor  # [B'] int (true prompt length)

    def __post_init__(self):
        if self.input_ids.dim() != 2:
            raise ValueError("input_ids must be 2-D")
        if self.attention_mask.shape != self.input_ids.shape:
            raise ValueError("attention_mask shape mismatch")
        if self.response_start_idx.shape != (self.input_ids.shape[0],):
            raise ValueError("response_start_idx must be [B']")

    @property
    def batch_size(self):
        return int(self.input_ids.shape[0])

    @property
    def seq_len(self):
        return int(self.input_ids.shape[1])


class RolloutEngine(ABC):
    name = "base"

    @abstractmethod
    def generate(self, request, sampling):
        """RolloutRequest + SamplingConfig -> RolloutBatch."""

    @abstractmethod
    def sync_weights(self, step):
        """Push updated weights to the rollout backend (no-op when
        co-located with training)."""

    def shutdown(self):
        return None


class HybridEngineRollout(RolloutEngine):
    """Co-located rollout over the training engine's hybrid generate.

    Prompts are grouped by true (unpadded) length; each group generates
    as one dense batch straight off the gathered training shards.
    """
    name = "hybrid_engine"

    def __init__(self, engine, pad_token_id=0, eos_token_id=None,
                 config=None):
        self.engine = engine
        self.pad_token_id = pad_token_id
        self.eos_token_id = eos_token_id
        self.config = config or RolloutConfig()

    def generate(self, request, sampling):
        ids, mask = request.prompt_ids, request.prompt_attention_mask
        n = sampling.n_samples_per_prompt
        lens = mask.sum(1).long()                      # true prompt lens
        B = ids.shape[0]
        rows = {}                                      # out_row -> (ids, plen)
        by_len = {}
        for b in range(B):
            by_len.setdefault(int(lens[b]), []).append(b)
        use_graph = (self.config.use_graph_capture
                     and sampling.temperature == 0
                     and torch.cuda.is_available())
        for plen, idxs in by_len.items():
            prompts = torch.stack(
                [ids[b, -plen:] for b in idxs]).repeat_interleave(n, 0)
            out = self.engine.generate(
                prompts, max_new_tokens=sampling.max_new_tokens,
                temperature=sampling.temperature,
                top_k=max(sampling.top_k, 0),
                top_p=sampling.top_p,
                eos_token_id=self.eos_token_id,
                use_hipgraph=use_graph)
            for j, b in enumerate(idxs):
                for s in range(n):
                    rows[b * n + s] = (out[j * n + s], plen)
        T = max(r.shape[0] for r, _ in rows.values())
        out_ids = torch.full((B * n, T), self.pad_token_id,
                             dtype=ids.dtype, device=ids.device)
        out_mask = torch.zeros(B * n, T, dtype=mask.dtype,
                               device=ids.device)
        starts = torch.zeros(B * n, dtype=torch.long, device=ids.device)
        for r, (seq, plen) in rows.items():
            out_ids[r, :seq.shape[0]] = seq
            out_mask[r, :seq.shape[0]] = 1
            starts[r] = plen
        return RolloutBatch(out_ids, out_mask, starts)

    def sync_weights(self, step):
        return None  # shares the training weights in place


def get_rollout_engine(engine, config=None, **kw):
    cfg = config or RolloutConfig()
    if cfg.engine != "hybrid_engine":
        raise ValueError(f"unknown rollout engine {cfg.engine!r}")
    return HybridEngineRollout(engine, config=cfg, **kw)
