"""InferenceEngine: KV-cache generation for the native model families.

Parity role: reference `deepspeed/inference/engine.py:43` (InferenceEngine)
/ `init_inference` (deepspeed/__init__.py:328). Round-1 scope: single-GPU
bf16 decode with preallocated KV caches and greedy/sampling generation;
TP sharding and ragged batching are the v2 follow-ups.
"""
import torch

from ..utils.logging import log_dist


class KVCache:
    """Preallocated [B, max_seq, Hk, D] cache for one layer."""

    def __init__(self, batch, max_seq, n_kv_heads, head_dim, dtype, device):
        self.k = torch.zeros(batch, max_seq, n_kv_heads, head_dim,
                             dtype=dtype, device=device)
        self.v = torch.zeros_like(self.k)
        self.pos = 0

    def update(self, k, v):
        """Append [B,S,Hk,D]; returns full views [B, pos+S, Hk, D]."""
        S = k.shape[1]
        self.k[:, self.pos:self.pos + S] = k
        self.v[:, self.pos:self.pos + S] = v
        self.pos += S
        return (self.k[:, :self.pos].contiguous(),
                self.v[:, :self.pos].contiguous())

    def reset(self):
        self.pos = 0


class InferenceConfig:
    def __init__(self, config=None, **kwargs):
        config = dict(config or {})
        config.update(kwargs)
        dt = config.get("dtype", torch.bfloat16)
        self.dtype = {"fp32": torch.float32, "float32": torch.float32,
                      "fp16": torch.float16, "half": torch.float16,
                      "float16": torch.float16, "bf16": torch.bfloat16,
                      "bfloat16": torch.bfloat16}.get(dt, dt) \
            if isinstance(dt, str) else dt
        self.max_out_tokens = config.get("max_out_tokens", 1024)
        self.tensor_parallel = config.get("tensor_parallel",
                                          {"tp_size": 1})
        self.replace_with_kernel_inject = config.get(
            "replace_with_kernel_inject", True)
        # int8 weight residency (ref init_inference dtype=torch.int8 +
        # quantization_setting): dequant-on-the-fly group-wise linears
        q = dict(config.get("quant", {}) or {})
        if self.dtype in (torch.int8, "int8"):
            q.setdefault("enabled", True)
            self.dtype = torch.bfloat16  # compute dtype stays bf16 MFMA
        self.quant = q


class InferenceEngine(torch.nn.Module):
    def __init__(self, model, config=None, **kwargs):
        super().__init__()
        self.module = model
        self._config = InferenceConfig(config, **kwargs)
        self.device = (torch.device("cuda", torch.cuda.current_device())
                       if torch.cuda.is_available()
                       else torch.device("cpu"))
        tp = self._config.tensor_parallel or {}
        tp_size = tp.get("tp_size", 1) if isinstance(tp, dict) \
            else getattr(tp, "tp_size", 1)
        if tp_size > 1:
            # shard attention/MLP linears across the TP group
            # (ref init_inference replace_with_kernel_inject + AutoTP)
            import torch.distributed as tdist
            from ..module_inject.auto_tp import apply_tensor_parallel
            assert tdist.is_initialized(), \
                "tensor_parallel.tp_size > 1 requires torch.distributed"
            group = tp.get("tp_group") if isinstance(tp, dict) else None
            apply_tensor_parallel(self.module, group)
        # HF models (config, no native .cfg): swap attention/norms/MLP
        # onto the HIP ops (ref replace_with_kernel_inject,
        # module_inject/replace_module.py:189)
        if self._config.replace_with_kernel_inject \
                and not hasattr(model, "cfg") \
                and hasattr(model, "config"):
            from ..module_inject.replace_module import \
                replace_transformer_layer
            replace_transformer_layer(self.module)
        self.module.to(self._config.dtype).to(self.device)
        if self._config.quant.get("enabled"):
            from .quantization import init_quantization
            q = self._config.quant
            init_quantization(
                self.module,
                quantize_bits=q.get("bits", 8),
                groups=q.get("groups", 64),
                mlp_extra_grouping=q.get("mlp_extra_grouping", False))
        self.module.eval()
        self._caches = None

    @property
    def _model_cfg(self):
        return getattr(self.module, "cfg", None)

    def _alloc_caches(self, batch, max_seq):
        cfg = self._model_cfg
        assert cfg is not None, "model must expose .cfg for KV caching"
        n_layers = cfg.num_hidden_layers
        self._caches = [KVCache(batch, max_seq, cfg.num_key_value_heads,
                                cfg.head_dim, self._config.dtype,
                                self.device)
                        for _ in range(n_layers)]

    @torch.no_grad()
    def forward(self, input_ids, **kwargs):
        return self.module(input_ids.to(self.device), **kwargs)

    @torch.no_grad()
    def generate_hipgraph(self, input_ids, max_new_tokens=32):
        """Greedy decode with the whole per-token step captured in ONE
        hipGraph (embed -> layers -> logits -> argmax -> feedback), so the
        decode loop is N graph replays with zero per-step launch overhead.

        MI355X-native answer to the reference's CUDA-graph decode
        (inference/engine.py:_create_cuda_graph) — 8B decode is otherwise
        launch-bound (~300 kernels/token).
        """
        from .graph_decode import hipgraph_greedy_decode
        return hipgraph_greedy_decode(self.module, self._model_cfg,
                                      self._config.dtype,
                                      input_ids.to(self.device),
                                      max_new_tokens)

    @torch.no_grad()
    def generate(self, input_ids, max_new_tokens=32, temperature=0.0,
                 top_k=0, eos_token_id=None, **hf_kwargs):
        """Greedy (temperature=0) or sampled generation with KV cache.

        Native models decode through the in-tree cache path; HF models
        (no .cfg) delegate to their own generate() with cache enabled.
        """
        input_ids = input_ids.to(self.device)
        if self._model_cfg is None and hasattr(self.module, "generate"):
            kw = dict(max_new_tokens=max_new_tokens,
                      do_sample=temperature > 0, **hf_kwargs)
            if temperature > 0:
                kw.update(temperature=temperature)
                if top_k:
                    kw.update(top_k=top_k)
            if eos_token_id is not None:
                kw.update(eos_token_id=eos_token_id)
            return self.module.generate(input_ids, **kw)
        B, S = input_ids.shape
        cfg = self._model_cfg
        max_seq = min(cfg.max_position_embeddings,
                      S + max_new_tokens)
        self._alloc_caches(B, max_seq)
        was_ckpt = getattr(cfg, "activation_checkpointing", False)
        cfg.activation_checkpointing = False
        try:
            # prefill
            logits = self.module(input_ids, kv_caches=self._caches)
            out = input_ids
            for step in range(max_new_tokens):
                next_logits = logits[:, -1, :].float()
                if temperature > 0:
                    next_logits = next_logits / temperature
                    if top_k > 0:
                        kth = torch.topk(next_logits, top_k,
                                         dim=-1).values[:, -1:]
                        next_logits = next_logits.masked_fill(
                            next_logits < kth, float("-inf"))
                    probs = torch.softmax(next_logits, dim=-1)
                    nxt = torch.multinomial(probs, 1)
                else:
                    nxt = next_logits.argmax(-1, keepdim=True)
                out = torch.cat([out, nxt], dim=1)
                if eos_token_id is not None and \
                        (nxt == eos_token_id).all():
                    break
                if out.shape[1] >= max_seq:
                    break
                logits = self.module(nxt, seq_offset=out.shape[1] - 1,
                                     kv_caches=self._caches)
            return out
        finally:
            cfg.activation_checkpointing = was_ckpt
            self._caches = None
