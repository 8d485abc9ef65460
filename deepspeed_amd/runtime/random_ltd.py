"""Random layerwise token dropping (random-LTD).

Parity: reference `runtime/data_pipeline/data_routing/basic_layer.py` +
`csrc/random_ltd/` gather/scatter kernels. Wrapped layers train on a
random token subset per step; kept tokens scatter back into the full
sequence (identity for dropped tokens). The token count follows a linear
schedule up to full length.
"""
import torch

from ..utils.logging import log_dist


class RandomLTDScheduler:
    def __init__(self, min_tokens, max_tokens, schedule_steps,
                 step_size=16):
        self.min_tokens = min_tokens
        self.max_tokens = max_tokens
        self.schedule_steps = max(1, schedule_steps)
        self.step_size = step_size
        self.current = min_tokens

    def update(self, global_step):
        frac = min(1.0, global_step / self.schedule_steps)
        n = self.min_tokens + frac * (self.max_tokens - self.min_tokens)
        n = int(n // self.step_size * self.step_size)
        self.current = max(self.min_tokens, min(n, self.max_tokens))
        return self.current

    def get_current_seq(self):
        return self.current

    def get_total_layer_tokens(self, train_iters):
        """Total tokens a wrapped layer processes over train_iters steps
        (ref data_routing/scheduler.py:55) — for budget accounting."""
        total = 0
        for it in range(int(train_iters)):
            frac = min(1.0, it / self.schedule_steps)
            n = self.min_tokens + frac * (self.max_tokens -
                                          self.min_tokens)
            total += int(n // self.step_size * self.step_size)
        return total

    def state_dict(self):
        return {"current": self.current}

    def load_state_dict(self, sd):
        self.current = sd["current"]


class RandomLTDLayer(torch.nn.Module):
    """Wraps a decoder layer: forward(x, *args) with token dropping.

    The wrapped layer must accept (x, cos, sin) like LlamaDecoderLayer;
    cos/sin rows are gathered for the kept positions so RoPE stays correct.
    """

    def __init__(self, layer, scheduler: RandomLTDScheduler):
        super().__init__()
        self.layer = layer
        self.scheduler = scheduler

    def forward(self, x, cos=None, sin=None, **kwargs):
        if not self.training:
            return self.layer(x, cos, sin, **kwargs)
        B, S = x.shape[0], x.shape[1]
        keep = min(self.scheduler.current, S)
        if keep >= S:
            return self.layer(x, cos, sin, **kwargs)
        idx = torch.sort(torch.randperm(S, device=x.device)[:keep]).values
        x_kept = x.index_select(1, idx)
        cos_k = cos.index_select(0, idx) if cos is not None else None
        sin_k = sin.index_select(0, idx) if sin is not None else None
        out_kept = self.layer(x_kept, cos_k, sin_k, **kwargs)
        out = x.clone()
        out.index_copy_(1, idx, out_kept)
        return out


def convert_to_random_ltd(model, layer_cls, min_tokens, max_tokens,
                          schedule_steps, layer_ids=None):
    """Wrap `layer_cls` submodules with random-LTD. `layer_ids` selects
    which occurrences (ref random_ltd_layer_id: usually the middle
    layers — first/last stay full-sequence)."""
    sched = RandomLTDScheduler(min_tokens, max_tokens, schedule_steps)
    targets = []
    for parent in model.modules():
        if isinstance(parent, RandomLTDLayer):
            continue
        for name, child in parent.named_children():
            if isinstance(child, layer_cls):
                targets.append((parent, name, child))
    n = 0
    for i, (parent, name, child) in enumerate(targets):
        if layer_ids is not None and i not in layer_ids:
            continue
        setattr(parent, name, RandomLTDLayer(child, sched))
        n += 1
    log_dist(f"random-LTD: wrapped {n} layers "
             f"({min_tokens}->{max_tokens} tokens)", ranks=[0])
    model.random_ltd_scheduler = sched
    return model
