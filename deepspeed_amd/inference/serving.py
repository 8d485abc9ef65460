"""Continuous batching serving engine (inference v2).

Parity: reference `inference/v2/ragged/*` + `mii` continuous batching:
requests enter and leave the running batch at token granularity — new
prompts prefill into free KV slots while in-flight sequences keep
decoding; every decode step runs ONE batched forward over all active
slots at their individual sequence lengths.

MI355X-native design: the KV pool is one tensor per layer
`[max_batch, max_seq, Hk, D]` (288 GB of HBM3E holds thousands of 8B
slots — paged/blocked KV is not needed to avoid fragmentation at this
scale, so slot granularity keeps the layout hipGraph-friendly). Ragged
lengths are handled with per-row RoPE positions (`positions` arg) and
an additive length mask through SDPA; the dedicated ragged-decode HIP
kernel (skinny GEMV + per-row KV scan) is tracked in ROADMAP.
"""
from dataclasses import dataclass, field
from typing import List, Optional

import torch


@dataclass
class Request:
    prompt: torch.Tensor                 # 1-D token ids
    max_new_tokens: int = 32
    eos_token_id: Optional[int] = None
    temperature: float = 0.0             # 0 = greedy
    top_k: int = 0
    top_p: float = 1.0
    rid: int = -1
    slot: int = -1
    generated: List[int] = field(default_factory=list)
    done: bool = False

    @property
    def tokens(self):
        return torch.cat([self.prompt,
                          torch.tensor(self.generated,
                                       dtype=self.prompt.dtype)])


class RaggedKVCache:
    """One layer's slot-granular KV pool with per-slot lengths."""

    def __init__(self, max_batch, max_seq, n_kv, d, dtype, device):
        self.k = torch.zeros(max_batch, max_seq, n_kv, d, dtype=dtype,
                             device=device)
        self.v = torch.zeros_like(self.k)
        self.lens = torch.zeros(max_batch, dtype=torch.long, device=device)
        self.rows = None        # active slot ids for this forward
        self.attn_mask = None
        self.ragged = None      # (kpool, vpool, rows, lens) for HIP decode
        self._prefill_slot = None
        self._prefill_off = 0   # chunked (SplitFuse-style) prefill offset
        # ragged decode HIP kernel reads straight from the pool — no
        # per-token gather copies (ref inference/v2 ragged_ops role)
        from ..ops.loader import get_ext
        self._use_ragged = (device is not None
                            and torch.device(device).type == "cuda"
                            and dtype == torch.bfloat16
                            and d in (64, 128)
                            and get_ext(required=False) is not None)

    # -- model-facing update ------------------------------------------------
    def update(self, k, v):
        if self._prefill_slot is not None:   # [1, S, Hk, D] prompt chunk
            s = self._prefill_slot
            S = k.shape[1]
            o = self._prefill_off
            self.k[s, o:o + S] = k[0]
            self.v[s, o:o + S] = v[0]
            self.lens[s] = o + S
            self.ragged = None
            if o == 0:
                self.attn_mask = None
                return k, v                  # causal prefill over itself
            # chunk at offset o attends to the cached prefix 0..o plus a
            # causal window within itself (Dynamic SplitFuse-style step)
            total = o + S
            q_pos = torch.arange(o, total, device=k.device).view(S, 1)
            k_pos = torch.arange(total, device=k.device).view(1, total)
            self.attn_mask = torch.where(k_pos <= q_pos, 0.0,
                                         float("-inf")) \
                .view(1, 1, S, total).to(torch.float32)
            return (self.k[s:s + 1, :total], self.v[s:s + 1, :total])
        rows = self.rows                     # decode: [n, 1, Hk, D]
        pos = self.lens[rows]
        self.k[rows, pos] = k[:, 0]
        self.v[rows, pos] = v[:, 0]
        self.lens[rows] = pos + 1
        lens = self.lens[rows]
        if self._use_ragged:
            self.attn_mask = None
            self.ragged = (self.k, self.v, rows.contiguous(),
                           lens.contiguous())
            return k, v  # attention reads from the pool via the kernel
        maxlen = int(lens.max())
        ar = torch.arange(maxlen, device=k.device)
        valid = ar.unsqueeze(0) < lens.unsqueeze(1)          # [n, maxlen]
        self.attn_mask = torch.where(
            valid, 0.0, float("-inf")).view(-1, 1, 1, maxlen).float()
        return self.k[rows, :maxlen], self.v[rows, :maxlen]


class ContinuousBatchingEngine:
    """Token-level continuous batching over a native model."""

    def __init__(self, model, max_batch=8, max_seq=None, device=None,
                 prefill_chunk=None, prefill_budget=None):
        self.model = model
        cfg = model.cfg if hasattr(model, "cfg") else model.config
        self.cfg = cfg
        self.device = device or next(model.parameters()).device
        self.max_seq = max_seq or cfg.max_position_embeddings
        self.max_batch = max_batch
        dtype = next(model.parameters()).dtype
        self.caches = [RaggedKVCache(max_batch, self.max_seq,
                                     cfg.num_key_value_heads, cfg.head_dim,
                                     dtype, self.device)
                       for _ in range(cfg.num_hidden_layers)]
        # Dynamic SplitFuse-style scheduling (ref FastGen): bound the
        # prompt tokens processed per step so long prompts stream in
        # chunks instead of stalling the decode batch. None = whole
        # prompt per step (legacy behavior). prefill_budget additionally
        # lets SEVERAL requests advance per step until the token budget
        # is spent (defaults to one chunk of one request).
        if prefill_budget and not prefill_chunk:
            prefill_chunk = prefill_budget  # budget implies chunking
        self.prefill_chunk = prefill_chunk
        self.prefill_budget = prefill_budget
        self.prefilling: List[Request] = []
        self.free_slots = list(range(max_batch))
        self.pending: List[Request] = []
        self.running: List[Request] = []
        self._next_rid = 0
        self._was_ckpt = getattr(cfg, "activation_checkpointing", False)
        cfg.activation_checkpointing = False

    def add_request(self, prompt, max_new_tokens=32, eos_token_id=None,
                    temperature=0.0, top_k=0, top_p=1.0):
        req = Request(prompt=prompt.to("cpu").long().view(-1),
                      max_new_tokens=max_new_tokens,
                      eos_token_id=eos_token_id, temperature=temperature,
                      top_k=top_k, top_p=top_p, rid=self._next_rid)
        self._next_rid += 1
        self.pending.append(req)
        return req.rid

    # -- internals ----------------------------------------------------------
    @torch.no_grad()
    def _prefill(self, req):
        if req.slot < 0:
            req.slot = self.free_slots.pop()
        for c in self.caches:
            c._prefill_slot = req.slot
        ids = req.prompt.view(1, -1).to(self.device)
        logits = self.model(ids, kv_caches=self.caches)
        for c in self.caches:
            c._prefill_slot = None
        req.generated.append(self._sample(logits[0, -1], req))
        self._check_done(req)

    @torch.no_grad()
    def _prefill_chunk_step(self, req):
        """Advance one request's prompt by at most prefill_chunk tokens;
        returns True when the prompt is fully ingested (and the first
        token sampled)."""
        pos = getattr(req, "prefill_pos", 0)
        n = len(req.prompt)
        take = min(self.prefill_chunk, n - pos)
        for c in self.caches:
            c._prefill_slot = req.slot
            c._prefill_off = pos
        ids = req.prompt[pos:pos + take].view(1, -1).to(self.device)
        logits = self.model(ids, kv_caches=self.caches, seq_offset=pos)
        for c in self.caches:
            c._prefill_slot = None
            c._prefill_off = 0
        req.prefill_pos = pos + take
        if req.prefill_pos >= n:
            req.generated.append(self._sample(logits[0, -1], req))
            self._check_done(req)
            return True
        return False

    @torch.no_grad()
    def _decode(self, active):
        rows = torch.tensor([r.slot for r in active], dtype=torch.long,
                            device=self.device)
        for c in self.caches:
            c.rows = rows
        last = torch.tensor([[r.generated[-1]] for r in active],
                            dtype=torch.long, device=self.device)
        positions = self.caches[0].lens[rows].view(-1, 1)
        logits = self.model(last, kv_caches=self.caches,
                            positions=positions)
        for c in self.caches:
            c.rows = None
        for r, row in zip(active, logits[:, -1]):
            r.generated.append(self._sample(row, r))
            self._check_done(r)

    @staticmethod
    def _sample(logits, req):
        logits = logits.float()
        if req.temperature <= 0:
            return int(logits.argmax())
        logits = logits / req.temperature
        if req.top_k > 0:
            kth = torch.topk(logits, req.top_k).values[-1]
            logits = logits.masked_fill(logits < kth, float("-inf"))
        probs = torch.softmax(logits, dim=-1)
        if req.top_p < 1.0:
            sp, idx = probs.sort(descending=True)
            cum = sp.cumsum(-1)
            keep = cum - sp < req.top_p  # keep until mass reaches top_p
            sp = sp * keep
            sp = sp / sp.sum()
            return int(idx[torch.multinomial(sp, 1)])
        return int(torch.multinomial(probs, 1))

    def _check_done(self, req):
        if len(req.generated) >= req.max_new_tokens or \
                (req.eos_token_id is not None
                 and req.generated[-1] == req.eos_token_id) or \
                len(req.prompt) + len(req.generated) >= self.max_seq:
            req.done = True

    # -- public loop --------------------------------------------------------
    @torch.no_grad()
    def step(self):
        """Admit + one decode step. Returns requests finished this step.

        With prefill_chunk set, at most ONE chunk of ONE prompt is
        ingested per step (bounded prefill work), interleaved with the
        decode batch — long prompts no longer stall running decodes."""
        if self.prefill_chunk is None:
            while self.pending and self.free_slots:
                req = self.pending.pop(0)
                self._prefill(req)
                self.running.append(req)
        else:
            while self.pending and self.free_slots:
                req = self.pending.pop(0)
                req.slot = self.free_slots.pop()
                req.prefill_pos = 0
                self.prefilling.append(req)
            budget = self.prefill_budget or self.prefill_chunk
            while self.prefilling and budget > 0:
                req = self.prefilling[0]
                remaining = len(req.prompt) - getattr(req, "prefill_pos", 0)
                budget -= min(self.prefill_chunk, remaining)
                if self._prefill_chunk_step(req):
                    self.prefilling.pop(0)
                    self.running.append(req)
                if self.prefill_budget is None:
                    break  # legacy: one chunk of one request per step
        active = [r for r in self.running if not r.done]
        if active:
            self._decode(active)
        finished = [r for r in self.running if r.done]
        for r in finished:
            self.free_slots.append(r.slot)
            self.running.remove(r)
        return finished

    def run(self, max_steps=10000):
        """Drain all pending/running requests; returns rid -> tokens."""
        out = {}
        steps = 0
        while (self.pending or self.running) and steps < max_steps:
            for r in self.step():
                out[r.rid] = r.tokens
            steps += 1
        return out

    def has_capacity(self):
        return bool(self.free_slots)

    def close(self):
        """Free the KV pool and restore the model's training config."""
        self.caches = []
        self.cfg.activation_checkpointing = self._was_ckpt

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()
        return False
