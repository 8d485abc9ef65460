"""Hybrid engine: RLHF train <-> generate switching on ZeRO-3 weights.

Parity: reference `runtime/hybrid_engine.py:31` (DeepSpeedHybridEngine).
`generate()` gathers the sharded parameters, runs KV-cache generation with
the training weights, then releases them back to shards — no weight copies,
no separate inference model.
"""
import contextlib

import torch

from ..inference.engine import KVCache
from ..utils.logging import log_dist


class HybridEngineMixin:
    """Mixed into DeepSpeedEngine (see engine.generate)."""


@contextlib.contextmanager
def gathered_for_generation(engine):
    """Gather all ZeRO-3 shards for the duration of generation.

    While everything is gathered, the per-submodule fetch/release hooks
    are paused: re-fetch/free per decode step is pure churn, and frees
    or collectives inside a hipGraph capture would poison the graph."""
    if engine.zero_optimization_stage() != 3:
        yield
        return
    from .zero.stage3_params import (ZeroParamStatus, all_gather_params,
                                     free_param, is_zero_param)
    params = [p for p in engine.module.parameters() if is_zero_param(p)]
    need = [p for p in params
            if p.ds_status == ZeroParamStatus.NOT_AVAILABLE]
    handle = all_gather_params(need, engine.optimizer.dp_group,
                               async_op=False)
    handle.wait()
    opt = engine.optimizer
    was_paused = getattr(opt, "_hooks_paused", False)
    opt._hooks_paused = True
    try:
        yield
    finally:
        opt._hooks_paused = was_paused
        for p in need:
            if not p.ds_persist:
                free_param(p)


@torch.no_grad()
def generate(engine, input_ids, max_new_tokens=32, temperature=0.0,
             top_k=0, top_p=1.0, eos_token_id=None,
             offload_states_during_generate=False, use_hipgraph=False):
    """Generate with the (possibly ZeRO-3-sharded) training weights.

    offload_states_during_generate: push optimizer masters/moments/grad
    accumulators to host for the generation phase (engine.offload_states)
    so long-rollout KV caches get the HBM — reloaded before returning.

    use_hipgraph: capture the per-token decode step as ONE hipGraph and
    replay it (ref runtime/hybrid_engine_graph.py role; greedy only —
    decode is launch-bound, ~300 kernels/token on 8B)."""
    if use_hipgraph and temperature > 0:
        raise ValueError("hipGraph rollout decode is greedy-only")
    if offload_states_during_generate:
        engine.offload_states(include=("hp_params", "lp_grads",
                                       "optim_states"))
    try:
        if use_hipgraph:
            return _generate_hipgraph(engine, input_ids, max_new_tokens)
        return _generate_inner(engine, input_ids, max_new_tokens,
                               temperature, top_k, top_p, eos_token_id)
    finally:
        if offload_states_during_generate:
            engine.reload_states()


@torch.no_grad()
def _generate_hipgraph(engine, input_ids, max_new_tokens):
    """Graph-captured greedy decode straight off the gathered training
    shards (the gather context holds every weight live across replays)."""
    from ..inference.graph_decode import hipgraph_greedy_decode
    module = engine.module
    cfg = getattr(module, "cfg", None)
    assert cfg is not None, "model must expose .cfg"
    was_training = module.training
    module.eval()
    try:
        with gathered_for_generation(engine):
            return hipgraph_greedy_decode(module, cfg, engine.config.dtype,
                                          input_ids, max_new_tokens)
    finally:
        module.train(was_training)


def _top_p_filter(logits, top_p):
    """Mask tokens outside the smallest nucleus with cumulative prob
    >= top_p (the highest-prob token always survives)."""
    srt, idx = torch.sort(logits, descending=True, dim=-1)
    cum = torch.softmax(srt, -1).cumsum(-1)
    drop_sorted = cum - torch.softmax(srt, -1) >= top_p
    drop = torch.zeros_like(drop_sorted).scatter(-1, idx, drop_sorted)
    return logits.masked_fill(drop, float("-inf"))


@torch.no_grad()
def _generate_inner(engine, input_ids, max_new_tokens, temperature,
                    top_k, top_p, eos_token_id):
    module = engine.module
    cfg = getattr(module, "cfg", None)
    assert cfg is not None, "model must expose .cfg"
    was_training = module.training
    was_ckpt = getattr(cfg, "activation_checkpointing", False)
    cfg.activation_checkpointing = False
    module.eval()
    try:
        with gathered_for_generation(engine):
            B, S = input_ids.shape
            max_seq = min(cfg.max_position_embeddings, S + max_new_tokens)
            caches = [KVCache(B, max_seq, cfg.num_key_value_heads,
                              cfg.head_dim, engine.config.dtype,
                              input_ids.device)
                      for _ in range(cfg.num_hidden_layers)]
            logits = module(input_ids, kv_caches=caches)
            out = input_ids
            for _ in range(max_new_tokens):
                nl = logits[:, -1, :].float()
                if temperature > 0:
                    nl = nl / temperature
                    if top_k > 0:
                        kth = torch.topk(nl, top_k, dim=-1).values[:, -1:]
                        nl = nl.masked_fill(nl < kth, float("-inf"))
                    if top_p < 1.0:
                        nl = _top_p_filter(nl, top_p)
                    nxt = torch.multinomial(torch.softmax(nl, -1), 1)
                else:
                    nxt = nl.argmax(-1, keepdim=True)
                out = torch.cat([out, nxt], dim=1)
                if eos_token_id is not None and (nxt == eos_token_id).all():
                    break
                if out.shape[1] >= max_seq:
                    break
                logits = module(nxt, seq_offset=out.shape[1] - 1,
                                kv_caches=caches)
            return out
    finally:
        cfg.activation_checkpointing = was_ckpt
        module.train(was_training)
