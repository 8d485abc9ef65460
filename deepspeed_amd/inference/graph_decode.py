"""Whole-step hipGraph greedy decode, shared by the inference engine
and the hybrid engine's rollout path.

Parity role: reference `inference/engine.py:_create_cuda_graph` and
`runtime/hybrid_engine_graph.py` (DecodeGraphCache). The MI355X design
differs: instead of one CUDA graph per decode position (the reference's
kernels read a host-side length counter frozen at capture), the ENTIRE
per-token step — embed -> layers -> logits -> argmax -> token feedback,
plus the position advance — runs on device tensors (StaticKVCache
indexes with device-side `index_copy_`), so ONE graph serves every
position and the decode loop is N replays with zero launch overhead.
"""
import torch


@torch.no_grad()
def hipgraph_greedy_decode(module, cfg, dtype, input_ids, max_new_tokens):
    """Greedy decode of `max_new_tokens` after an eager prefill; the
    per-token step is captured once and replayed. Requires a GPU and a
    native model exposing `.model.rope_cos/rope_sin` (llama family)."""
    from ..models.llama import StaticKVCache, llama_decode_step
    assert torch.cuda.is_available(), "hipGraph decode needs a GPU"
    device = input_ids.device
    B, S = input_ids.shape
    max_seq = min(cfg.max_position_embeddings, S + max_new_tokens)
    caches = [StaticKVCache(B, max_seq, cfg.num_key_value_heads,
                            cfg.head_dim, dtype, device)
              for _ in range(cfg.num_hidden_layers)]
    was_ckpt = cfg.activation_checkpointing
    cfg.activation_checkpointing = False
    try:
        # ---- prefill (eager) ----
        class _Adapter:
            def __init__(self, sc):
                self.sc = sc

            def update(self, k, v):
                self.sc.prefill(k, v)
                n = k.shape[1]
                return (self.sc.k[:, :n].contiguous(),
                        self.sc.v[:, :n].contiguous())

        adapters = [_Adapter(c) for c in caches]
        logits = module(input_ids, kv_caches=adapters)

        # ---- static state ----
        cos_t = module.model.rope_cos.float().to(device)
        sin_t = module.model.rope_sin.float().to(device)
        pos_idx = torch.tensor([S - 1], device=device)
        attn_mask = torch.full((1, 1, 1, max_seq), float("-inf"),
                               device=device, dtype=dtype)
        attn_mask[..., :S] = 0.0
        id_buf = logits[:, -1, :].argmax(-1, keepdim=True)
        out_tokens = torch.zeros(B, max_new_tokens, dtype=torch.long,
                                 device=device)
        step_idx = torch.zeros(1, dtype=torch.long, device=device)

        def one_step():
            out_tokens.index_copy_(1, step_idx, id_buf)
            step_idx.add_(1)
            pos_idx.add_(1)
            attn_mask.index_fill_(3, pos_idx, 0.0)
            lg = llama_decode_step(module, id_buf, caches, pos_idx,
                                   attn_mask, cos_t, sin_t)
            id_buf.copy_(lg[:, -1, :].argmax(-1, keepdim=True))

        # warmup on a side stream (allocator settles), then capture
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            one_step()
        torch.cuda.current_stream().wait_stream(s)
        n_graphed = max_new_tokens - 1
        if n_graphed > 0:
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph):
                one_step()
            for _ in range(n_graphed - 1):
                graph.replay()
        out_tokens.index_copy_(1, step_idx, id_buf)
        return torch.cat([input_ids, out_tokens], dim=1)
    finally:
        cfg.activation_checkpointing = was_ckpt
