"""hipGraph training-step compilation (DeepCompile-runtime equivalent).

Parity role: reference DeepCompile (`deepspeed/compile/*`) builds a
torch.compile backend that fuses ZeRO comm scheduling into the captured
graph. On ROCm there is no Triton/inductor codegen in this stack, and the
hot compute is already hand-written HIP — what remains on the table is
LAUNCH OVERHEAD and host gaps on fixed-shape training steps. The
MI355X-native answer is hipGraph capture:

- `graph_compile(module, sample)` wraps the module's forward+backward in
  hipGraphs via `torch.cuda.make_graphed_callables`. Only the module's
  INTERNAL compute is captured; gradient accumulation into `p.grad`
  (views into the ZeRO flat buffers) and the post-accumulate hooks that
  schedule bucketed RCCL reduce-scatters stay eager — so ZeRO-1/2 comm
  and overlap behave exactly as uncompiled.
- `DeepSpeedEngine.compile(sample_input=...)` applies it in place.

Constraints (checked): CUDA available, static input shapes, ZeRO stage
< 3 (stage 3's per-module gather/release rewrites parameter storage
mid-forward, which a captured graph cannot see), no dropout-style
host-side RNG branching in the module.

The inference analogue (whole-decode-step capture incl. cache update and
argmax feedback) lives in `inference/engine.py:generate_hipgraph`.
"""
import torch

from .utils.logging import log_dist


class _PositionalLM(torch.nn.Module):
    """Graph capture takes positional args only; adapt (ids, labels)."""

    def __init__(self, mod):
        super().__init__()
        self.mod = mod

    def forward(self, input_ids, labels):
        return self.mod(input_ids, labels=labels)


class _GraphedLM(torch.nn.Module):
    """Training calls replay the captured graphs; label-free calls
    (generation), extra kwargs, or inputs whose SHAPE differs from the
    captured static shape fall back to the eager module (replaying a
    hipGraph on mismatched shapes reads stale capture buffers — silent
    corruption, not an error)."""

    def __init__(self, graphed, eager, static_shape, static_label_shape):
        super().__init__()
        self.graphed = graphed
        self.eager = eager
        self._shape = tuple(static_shape)
        self._lshape = tuple(static_label_shape) \
            if static_label_shape is not None else None
        self._warned = False

    def forward(self, input_ids, labels=None, **kw):
        if labels is not None and not kw \
                and tuple(input_ids.shape) == self._shape \
                and (self._lshape is None
                     or tuple(labels.shape) == self._lshape):
            return self.graphed(input_ids, labels)
        if labels is not None and not self._warned:
            self._warned = True
            log_dist(
                f"hipGraph fallback to eager: input {tuple(input_ids.shape)}"
                f" != captured {self._shape}", ranks=[0])
        return self.eager(input_ids, labels=labels, **kw)


def graph_compile(module, sample_input, sample_labels=None,
                  num_warmup_iters=3):
    """Return `module` with fwd+bwd captured as hipGraphs.

    sample_input (and sample_labels for loss-computing LMs) must carry
    the static training shapes; the module must already be on the GPU in
    its training dtype.
    """
    if not torch.cuda.is_available():
        raise RuntimeError("graph_compile requires a GPU (hipGraph capture)")
    dev = next(module.parameters()).device
    sample = sample_input.to(dev)
    if sample_labels is not None:
        lbl = sample_labels.to(dev)
        graphed = torch.cuda.make_graphed_callables(
            _PositionalLM(module), (sample, lbl),
            num_warmup_iters=num_warmup_iters)
        out = _GraphedLM(graphed, module, sample.shape, lbl.shape)
    else:
        out = torch.cuda.make_graphed_callables(
            module, (sample,), num_warmup_iters=num_warmup_iters)
    log_dist("hipGraph-compiled module forward+backward "
             f"(input {tuple(sample.shape)})", ranks=[0])
    return out


def engine_compile(engine, sample_input, sample_labels=None,
                   num_warmup_iters=3):
    """In-place `engine.compile()`: swap the engine's module for the
    graphed one. ZeRO stage must be < 3."""
    stage = engine.zero_optimization_stage()
    if stage >= 3:
        raise RuntimeError(
            "hipGraph step capture is incompatible with ZeRO-3's dynamic "
            "parameter gathering; use stage <= 2 (or rely on ZeRO-3's "
            "stream overlap, which already hides launch gaps)")
    engine.module = graph_compile(engine.module, sample_input,
                                  sample_labels, num_warmup_iters)
    return engine
